"""Load real Stable Diffusion checkpoints (ldm / sdwui naming) into the
native modules.

Users of the reference extension point every worker at the same sdwui
checkpoint file (ref world.py:784-811 sync + shared.py model options).
Here the equivalent is: hand ``load_ldm_state_dict`` the state dict of an
SD1.5-lineage ``.safetensors``/``.ckpt`` (``model.diffusion_model.*``,
``first_stage_model.*``, ``cond_stage_model.transformer.*``) and it fills
the native UNet/VAE/CLIP, adapting the layout differences:

- spatial-transformer ``proj_in``/``proj_out`` and VAE attention q/k/v are
  1x1 convs in SD1.5 files but linears here (squeezed on load, expanded on
  export);
- CLIP stores separate q/k/v projections; the native encoder runs one fused
  qkv linear (concatenated on load, split on export);
- the file's ``quant_conv``/``post_quant_conv`` 1x1 convs have no native
  module — they are folded exactly into ``encoder.conv_out`` /
  ``decoder.conv_in`` (a 1x1 conv composed with a conv is still a conv).

``to_ldm_state_dict`` is the inverse (identity quant convs), so native
checkpoints can be exported for sdwui tooling and the pair round-trips
bit-exactly at fp32.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from ..utils import get_logger

log = get_logger("sdwd.convert")

UNET_PREFIX = "model.diffusion_model."
VAE_PREFIX = "first_stage_model."
CLIP_PREFIX = "cond_stage_model.transformer.text_model."
# SDXL checkpoints: two text encoders under the sgm conditioner —
# embedders.0 is SD-style CLIP-L, embedders.1 is open_clip (bigG) layout
XL_CLIP_L_PREFIX = "conditioner.embedders.0.transformer.text_model."
XL_CLIP_G_PREFIX = "conditioner.embedders.1.model."
# the SDXL REFINER has no CLIP-L: its CLIP-G sits at embedder index 0
XL_REFINER_G_PREFIX = "conditioner.embedders.0.model."
# SD2.x: single open_clip (ViT-H) text tower
SD2_CLIP_PREFIX = "cond_stage_model.model."

_RES_INNER = {
    "norm1": "in_layers.0",
    "conv1": "in_layers.2",
    "emb_proj": "emb_layers.1",
    "norm2": "out_layers.0",
    "conv2": "out_layers.3",
    "skip": "skip_connection",
}

_VAE_RES_INNER = {
    "norm1": "norm1",
    "conv1": "conv1",
    "norm2": "norm2",
    "conv2": "conv2",
    "skip": "nin_shortcut",
}


def _map_inner(params: List[str], table: Dict[str, str], ours: str, ldm: str,
               out: Dict[str, str]) -> None:
    for p in params:
        head, _, tail = p.partition(".")
        out[f"{ldm}.{table[head]}.{tail}"] = f"{ours}.{p}"


def _transformer_map(st, ours: str, ldm: str, out: Dict[str, str]) -> None:
    """SpatialTransformer: ldm names differ in to_out.0 / ff.net.*."""
    for name in ("norm", "proj_in", "proj_out"):
        for p in ("weight", "bias"):
            out[f"{ldm}.{name}.{p}"] = f"{ours}.{name}.{p}"
    for i, blk in enumerate(st.blocks):
        o = f"{ours}.blocks.{i}"
        l = f"{ldm}.transformer_blocks.{i}"
        for n in ("norm1", "norm2", "norm3"):
            out[f"{l}.{n}.weight"] = f"{o}.{n}.weight"
            out[f"{l}.{n}.bias"] = f"{o}.{n}.bias"
        for a in ("attn1", "attn2"):
            for n in ("to_q", "to_k", "to_v"):
                out[f"{l}.{a}.{n}.weight"] = f"{o}.{a}.{n}.weight"
            out[f"{l}.{a}.to_out.0.weight"] = f"{o}.{a}.to_out.weight"
            out[f"{l}.{a}.to_out.0.bias"] = f"{o}.{a}.to_out.bias"
        out[f"{l}.ff.net.0.proj.weight"] = f"{o}.ff.0.proj.weight"
        out[f"{l}.ff.net.0.proj.bias"] = f"{o}.ff.0.proj.bias"
        out[f"{l}.ff.net.2.weight"] = f"{o}.ff.1.weight"
        out[f"{l}.ff.net.2.bias"] = f"{o}.ff.1.bias"


def _seq_map(seq, ours: str, ldm: str,
             out: Dict[str, str]) -> None:
    from .unet import Downsample, ResBlock, SpatialTransformer, Upsample

    for m, mod in enumerate(seq.mods):
        o = f"{ours}.mods.{m}"
        l = f"{ldm}.{m}"
        if isinstance(mod, ResBlock):
            params = [k for k, _ in mod.named_parameters()]
            _map_inner(params, _RES_INNER, o, l, out)
        elif isinstance(mod, SpatialTransformer):
            _transformer_map(mod, o, l, out)
        elif isinstance(mod, Downsample):
            out[f"{l}.op.weight"] = f"{o}.conv.weight"
            out[f"{l}.op.bias"] = f"{o}.conv.bias"
        elif isinstance(mod, Upsample):
            out[f"{l}.conv.weight"] = f"{o}.conv.weight"
            out[f"{l}.conv.bias"] = f"{o}.conv.bias"


def unet_key_map(unet) -> Dict[str, str]:
    """-> {ldm_key (no prefix): native_key} for this UNet's config."""
    out: Dict[str, str] = {}
    for a, b in (("0", "0"), ("2", "2")):
        for p in ("weight", "bias"):
            out[f"time_embed.{a}.{p}"] = f"time_mlp.{b}.{p}"
    if unet.label_mlp is not None:
        for a in ("0", "2"):
            for p in ("weight", "bias"):
                out[f"label_emb.0.{a}.{p}"] = f"label_mlp.{a}.{p}"
    for p in ("weight", "bias"):
        out[f"input_blocks.0.0.{p}"] = f"conv_in.{p}"
        out[f"out.0.{p}"] = f"norm_out.{p}"
        out[f"out.2.{p}"] = f"conv_out.{p}"
    for n, seq in enumerate(unet.down):
        _seq_map(seq, f"down.{n}", f"input_blocks.{n + 1}", out)
    _seq_map(unet.mid, "mid", "middle_block", out)
    for n, seq in enumerate(unet.up):
        _seq_map(seq, f"up.{n}", f"output_blocks.{n}", out)
    return out


def vae_key_map(vae) -> Dict[str, str]:
    """-> {ldm_key (no prefix): native_key}. quant convs handled separately."""
    cfg = vae.cfg
    out: Dict[str, str] = {}
    levels = len(cfg.channel_mult)

    def res(ours: str, ldm: str, mod) -> None:
        params = [k for k, _ in mod.named_parameters()]
        _map_inner(params, _VAE_RES_INNER, ours, ldm, out)

    def attn(ours: str, ldm: str) -> None:
        for p in ("weight", "bias"):
            out[f"{ldm}.norm.{p}"] = f"{ours}.norm.{p}"
            out[f"{ldm}.q.{p}"] = f"{ours}.q.{p}"
            out[f"{ldm}.k.{p}"] = f"{ours}.k.{p}"
            out[f"{ldm}.v.{p}"] = f"{ours}.v.{p}"
            out[f"{ldm}.proj_out.{p}"] = f"{ours}.out.{p}"

    # encoder: flat blocks list <-> down.{lvl}.block/downsample
    idx = 0
    for lvl in range(levels):
        for j in range(cfg.num_res_blocks):
            res(f"encoder.blocks.{idx}", f"encoder.down.{lvl}.block.{j}",
                vae.encoder.blocks[idx])
            idx += 1
        if lvl != levels - 1:
            for p in ("weight", "bias"):
                out[f"encoder.down.{lvl}.downsample.conv.{p}"] = (
                    f"encoder.blocks.{idx}.conv.{p}"
                )
            idx += 1
    res("encoder.mid.0", "encoder.mid.block_1", vae.encoder.mid[0])
    attn("encoder.mid.1", "encoder.mid.attn_1")
    res("encoder.mid.2", "encoder.mid.block_2", vae.encoder.mid[2])
    # decoder: flat list runs highest level first; ldm stores that as
    # up.{levels-1} (execution order is reversed storage order)
    res("decoder.mid.0", "decoder.mid.block_1", vae.decoder.mid[0])
    attn("decoder.mid.1", "decoder.mid.attn_1")
    res("decoder.mid.2", "decoder.mid.block_2", vae.decoder.mid[2])
    idx = 0
    for lvl in reversed(range(levels)):
        for j in range(cfg.num_res_blocks + 1):
            res(f"decoder.blocks.{idx}", f"decoder.up.{lvl}.block.{j}",
                vae.decoder.blocks[idx])
            idx += 1
        if lvl != 0:
            for p in ("weight", "bias"):
                out[f"decoder.up.{lvl}.upsample.conv.{p}"] = (
                    f"decoder.blocks.{idx}.conv.{p}"
                )
            idx += 1
    for part in ("encoder", "decoder"):
        for name in ("conv_in", "conv_out", "norm_out"):
            for p in ("weight", "bias"):
                out[f"{part}.{name}.{p}"] = f"{part}.{name}.{p}"
    return out


def clip_key_map(enc) -> Tuple[Dict[str, str], List[int]]:
    """-> ({ldm_key: native_key} for the 1:1 part, layer indices whose
    q/k/v projections must be fused into ``blocks.{i}.attn.qkv``)."""
    out = {
        "embeddings.token_embedding.weight": "token_emb.weight",
        "embeddings.position_embedding.weight": "pos_emb",
        "final_layer_norm.weight": "ln_final.weight",
        "final_layer_norm.bias": "ln_final.bias",
    }
    fused = []
    for i in range(len(enc.blocks)):
        l = f"encoder.layers.{i}"
        o = f"blocks.{i}"
        for a, b in (("layer_norm1", "ln1"), ("layer_norm2", "ln2")):
            out[f"{l}.{a}.weight"] = f"{o}.{b}.weight"
            out[f"{l}.{a}.bias"] = f"{o}.{b}.bias"
        out[f"{l}.self_attn.out_proj.weight"] = f"{o}.attn.out.weight"
        out[f"{l}.self_attn.out_proj.bias"] = f"{o}.attn.out.bias"
        out[f"{l}.mlp.fc1.weight"] = f"{o}.mlp.0.weight"
        out[f"{l}.mlp.fc1.bias"] = f"{o}.mlp.0.bias"
        out[f"{l}.mlp.fc2.weight"] = f"{o}.mlp.2.weight"
        out[f"{l}.mlp.fc2.bias"] = f"{o}.mlp.2.bias"
        fused.append(i)
    return out, fused


def openclip_key_map(enc) -> Dict[str, str]:
    """open_clip text-tower layout (SDXL's second encoder). qkv is stored
    fused (attn.in_proj_*) which matches the native fused qkv directly."""
    out = {
        "token_embedding.weight": "token_emb.weight",
        "positional_embedding": "pos_emb",
        "ln_final.weight": "ln_final.weight",
        "ln_final.bias": "ln_final.bias",
    }
    for i in range(len(enc.blocks)):
        l = f"transformer.resblocks.{i}"
        o = f"blocks.{i}"
        for a, b in (("ln_1", "ln1"), ("ln_2", "ln2")):
            out[f"{l}.{a}.weight"] = f"{o}.{b}.weight"
            out[f"{l}.{a}.bias"] = f"{o}.{b}.bias"
        out[f"{l}.attn.in_proj_weight"] = f"{o}.attn.qkv.weight"
        out[f"{l}.attn.in_proj_bias"] = f"{o}.attn.qkv.bias"
        out[f"{l}.attn.out_proj.weight"] = f"{o}.attn.out.weight"
        out[f"{l}.attn.out_proj.bias"] = f"{o}.attn.out.bias"
        out[f"{l}.mlp.c_fc.weight"] = f"{o}.mlp.0.weight"
        out[f"{l}.mlp.c_fc.bias"] = f"{o}.mlp.0.bias"
        out[f"{l}.mlp.c_proj.weight"] = f"{o}.mlp.2.weight"
        out[f"{l}.mlp.c_proj.bias"] = f"{o}.mlp.2.bias"
    return out


def _fit(src: torch.Tensor, like: torch.Tensor) -> torch.Tensor:
    """Adapt a file tensor to the native parameter's shape: squeeze 1x1
    convs stored for linears and vice versa."""
    if src.shape == like.shape:
        return src
    if src.dim() == 4 and like.dim() == 2 and src.shape[2:] == (1, 1):
        return src.reshape(src.shape[:2])
    if src.dim() == 2 and like.dim() == 4 and like.shape[2:] == (1, 1):
        return src.reshape(*src.shape, 1, 1)
    raise ValueError(f"shape mismatch {tuple(src.shape)} vs {tuple(like.shape)}")


def _fold_output_1x1(conv_w, conv_b, q_w, q_b):
    """y = Q(conv(x)) with Q a 1x1 conv -> one conv: W' = Q.W @ W, b' = Q(b)."""
    q = q_w.reshape(q_w.shape[0], q_w.shape[1]).to(torch.float64)
    w = conv_w.to(torch.float64)
    new_w = torch.einsum("oc,cikl->oikl", q, w)
    new_b = q @ conv_b.to(torch.float64) + q_b.to(torch.float64)
    return new_w.to(conv_w.dtype), new_b.to(conv_b.dtype)


def _fold_input_1x1(conv_w, conv_b, q_w, q_b):
    """y = conv(Q(x)) with Q a 1x1 conv -> one conv: W'[:,c] = sum_m W[:,m] Q[m,c],
    b' = b + sum_{m,kh,kw} W[o,m,kh,kw] * Q.b[m]."""
    q = q_w.reshape(q_w.shape[0], q_w.shape[1]).to(torch.float64)
    w = conv_w.to(torch.float64)
    new_w = torch.einsum("omkl,mc->ockl", w, q)
    new_b = conv_b.to(torch.float64) + torch.einsum(
        "omkl,m->o", w, q_b.to(torch.float64)
    )
    return new_w.to(conv_w.dtype), new_b.to(conv_b.dtype)


def _load_part(mod, sub: Dict[str, torch.Tensor], key_map: Dict[str, str],
               report: Dict[str, list]) -> None:
    own = dict(mod.named_parameters())
    own.update(dict(mod.named_buffers()))
    with torch.no_grad():
        for ldm_key, our_key in key_map.items():
            if ldm_key not in sub:
                report["missing"].append(ldm_key)
                continue
            tgt = own.get(our_key)
            if tgt is None:
                report["missing"].append(our_key)
                continue
            tgt.copy_(_fit(sub.pop(ldm_key), tgt).to(tgt.dtype))
            report["loaded"].append(ldm_key)


def _load_sd_clip(enc, clip_sd: Dict[str, torch.Tensor],
                  report: Dict[str, list]) -> None:
    """SD-style CLIP (split q/k/v projections) -> native fused qkv."""
    kmap, fused = clip_key_map(enc)
    _load_part(enc, clip_sd, kmap, report)
    with torch.no_grad():
        for i in fused:
            parts_w, parts_b = [], []
            ok = True
            for n in ("q_proj", "k_proj", "v_proj"):
                kw = f"encoder.layers.{i}.self_attn.{n}.weight"
                kb = f"encoder.layers.{i}.self_attn.{n}.bias"
                if kw not in clip_sd or kb not in clip_sd:
                    ok = False
                    report["missing"].append(kw)
                    continue
                parts_w.append(clip_sd.pop(kw))
                parts_b.append(clip_sd.pop(kb))
            if ok:
                qkv = enc.blocks[i].attn.qkv
                qkv.weight.copy_(torch.cat(parts_w, 0).to(qkv.weight.dtype))
                qkv.bias.copy_(torch.cat(parts_b, 0).to(qkv.bias.dtype))
                report["loaded"].append(f"encoder.layers.{i}.self_attn.qkv")


def _load_vae_part(vae, vae_sd: Dict[str, torch.Tensor],
                   report: Dict[str, list]) -> None:
    """Fill ``vae`` from bare ldm VAE keys (no first_stage_model. prefix),
    folding the file's quant_conv/post_quant_conv 1x1s into the native
    encoder.conv_out / decoder.conv_in (this module's docstring).
    Consumed keys are popped from ``vae_sd``."""
    quant = {k: vae_sd.pop(k) for k in list(vae_sd) if "quant_conv" in k}
    _load_part(vae, vae_sd, vae_key_map(vae), report)
    with torch.no_grad():
        enc_out = dict(vae.encoder.named_parameters())
        if "quant_conv.weight" in quant:
            dev = enc_out["conv_out.weight"].device
            w, b = _fold_output_1x1(
                enc_out["conv_out.weight"].float(),
                enc_out["conv_out.bias"].float(),
                quant["quant_conv.weight"].float().to(dev),
                quant["quant_conv.bias"].float().to(dev),
            )
            enc_out["conv_out.weight"].copy_(w.to(enc_out["conv_out.weight"].dtype))
            enc_out["conv_out.bias"].copy_(b.to(enc_out["conv_out.bias"].dtype))
        dec = dict(vae.decoder.named_parameters())
        if "post_quant_conv.weight" in quant:
            dev = dec["conv_in.weight"].device
            w, b = _fold_input_1x1(
                dec["conv_in.weight"].float(),
                dec["conv_in.bias"].float(),
                quant["post_quant_conv.weight"].float().to(dev),
                quant["post_quant_conv.bias"].float().to(dev),
            )
            dec["conv_in.weight"].copy_(w.to(dec["conv_in.weight"].dtype))
            dec["conv_in.bias"].copy_(b.to(dec["conv_in.bias"].dtype))


def load_vae_state_dict(vae, state: Dict[str, torch.Tensor]
                        ) -> Dict[str, list]:
    """Load a STANDALONE VAE file (the sdwui "SD VAE" dropdown files the
    reference synced by name through load_options — ref worker.py:646-688):
    keys are either ``first_stage_model.*`` (as in full checkpoints) or the
    bare ldm VAE naming used by published vae-ft-* / anime VAE files.
    EMA/loss bookkeeping keys are tolerated."""
    report: Dict[str, list] = {"loaded": [], "missing": [], "unexpected": []}
    if any(k.startswith(VAE_PREFIX) for k in state):
        sub = {
            k[len(VAE_PREFIX):]: v
            for k, v in state.items() if k.startswith(VAE_PREFIX)
        }
    else:
        sub = {
            k: v for k, v in state.items()
            if not k.startswith(("loss.", "model_ema.", "ema."))
        }
    _load_vae_part(vae, sub, report)
    report["unexpected"] += list(sub)
    return report


def load_ldm_state_dict(bundle, state: Dict[str, torch.Tensor]) -> Dict[str, list]:
    """Fill ``bundle`` (unet/vae/text_encoder) from an sdwui/ldm state dict.

    Returns {"loaded": [...], "missing": [...], "unexpected": [...]} so the
    caller can decide whether a partial load is acceptable.
    """
    report: Dict[str, list] = {"loaded": [], "missing": [], "unexpected": []}
    state = dict(state)

    def take(prefix: str) -> Dict[str, torch.Tensor]:
        sub = {}
        for k in list(state):
            if k.startswith(prefix):
                sub[k[len(prefix):]] = state.pop(k)
        return sub

    unet_sd = take(UNET_PREFIX)
    vae_sd = take(VAE_PREFIX)
    clip_sd = take(CLIP_PREFIX)

    if unet_sd:
        _load_part(bundle.unet, unet_sd, unet_key_map(bundle.unet), report)
        report["unexpected"] += [UNET_PREFIX + k for k in unet_sd]
    if vae_sd:
        _load_vae_part(bundle.vae, vae_sd, report)
        report["unexpected"] += [VAE_PREFIX + k for k in vae_sd]
    if clip_sd and bundle.text_encoder is not None:
        _load_sd_clip(bundle.text_encoder, clip_sd, report)
        report["unexpected"] += [CLIP_PREFIX + k for k in clip_sd]
    sd2_clip = take(SD2_CLIP_PREFIX)
    if sd2_clip and bundle.text_encoder is not None:
        enc = bundle.text_encoder
        _load_part(enc, sd2_clip, openclip_key_map(enc), report)
        if "text_projection" in sd2_clip:
            enc.set_text_projection(sd2_clip.pop("text_projection").float())
            report["loaded"].append("text_projection")
        sd2_clip.pop("logit_scale", None)
        report["unexpected"] += [SD2_CLIP_PREFIX + k for k in sd2_clip]
    xl_l = take(XL_CLIP_L_PREFIX)
    if xl_l and bundle.text_encoder is not None:
        _load_sd_clip(bundle.text_encoder, xl_l, report)
        report["unexpected"] += [XL_CLIP_L_PREFIX + k for k in xl_l]
    xl_g = take(XL_CLIP_G_PREFIX)
    if not xl_g and getattr(bundle, "is_refiner", False):
        xl_g = take(XL_REFINER_G_PREFIX)
    if xl_g and getattr(bundle, "text_encoder_2", None) is not None:
        enc2 = bundle.text_encoder_2
        _load_part(enc2, xl_g, openclip_key_map(enc2), report)
        if "text_projection" in xl_g:
            enc2.set_text_projection(xl_g.pop("text_projection").float())
            report["loaded"].append("text_projection")
        xl_g.pop("logit_scale", None)
        report["unexpected"] += [XL_CLIP_G_PREFIX + k for k in xl_g]
    # whatever is left in the file we never claimed to consume
    report["unexpected"] += [
        k for k in state
        if not k.startswith(("cond_stage_model.", "conditioner.", "model_ema.",
                             "alphas", "betas", "sqrt_", "log_one_minus",
                             "posterior_", "model.", "first_stage_model."))
    ]
    if report["missing"]:
        log.warning("ldm load: %d keys missing (first: %s)",
                    len(report["missing"]), report["missing"][:3])
    log.info("ldm load: %d tensors loaded, %d unexpected",
             len(report["loaded"]), len(report["unexpected"]))
    return report


def to_ldm_state_dict(bundle) -> Dict[str, torch.Tensor]:
    """Export the bundle under sdwui/ldm naming (SD1.5 conventions: 1x1-conv
    proj_in/proj_out and VAE attention tensors, identity quant convs, split
    CLIP q/k/v)."""
    out: Dict[str, torch.Tensor] = {}
    unet_sd = bundle.unet.state_dict()
    for ldm_key, our_key in unet_key_map(bundle.unet).items():
        t = unet_sd[our_key]
        if ("proj_in.weight" in ldm_key or "proj_out.weight" in ldm_key) and t.dim() == 2:
            t = t.reshape(*t.shape, 1, 1)
        out[UNET_PREFIX + ldm_key] = t
    vae_sd = bundle.vae.state_dict()
    for ldm_key, our_key in vae_key_map(bundle.vae).items():
        t = vae_sd[our_key]
        if (".attn_1." in ldm_key and "norm" not in ldm_key
                and t.dim() == 2):
            t = t.reshape(*t.shape, 1, 1)
        out[VAE_PREFIX + ldm_key] = t
    lat = 2 * bundle.vae.cfg.latent_channels
    eye = torch.eye(lat, dtype=vae_sd["encoder.conv_out.weight"].dtype)
    out[VAE_PREFIX + "quant_conv.weight"] = eye.reshape(lat, lat, 1, 1)
    out[VAE_PREFIX + "quant_conv.bias"] = torch.zeros(lat, dtype=eye.dtype)
    lat //= 2
    eye = torch.eye(lat, dtype=eye.dtype)
    out[VAE_PREFIX + "post_quant_conv.weight"] = eye.reshape(lat, lat, 1, 1)
    out[VAE_PREFIX + "post_quant_conv.bias"] = torch.zeros(lat, dtype=eye.dtype)
    clip_l_prefix = XL_CLIP_L_PREFIX if bundle.is_sdxl else CLIP_PREFIX
    # SD2.x lineage: single open_clip tower (d_model 1024 is unambiguous —
    # SD1.x text encoders are always 768-wide HF-CLIP)
    if (not bundle.is_sdxl and bundle.text_encoder is not None
            and bundle.text_encoder.d_model == 1024):
        enc = bundle.text_encoder
        sd2 = enc.state_dict()
        for ldm_key, our_key in openclip_key_map(enc).items():
            out[SD2_CLIP_PREFIX + ldm_key] = sd2[our_key]
        tp = sd2.get("text_proj")
        out[SD2_CLIP_PREFIX + "text_projection"] = (
            tp if tp is not None else torch.eye(enc.d_model)
        )
        return out
    if bundle.text_encoder is not None:
        enc = bundle.text_encoder
        clip_sd = enc.state_dict()
        kmap, fused = clip_key_map(enc)
        for ldm_key, our_key in kmap.items():
            out[clip_l_prefix + ldm_key] = clip_sd[our_key]
        for i in fused:
            w = clip_sd[f"blocks.{i}.attn.qkv.weight"].chunk(3, 0)
            b = clip_sd[f"blocks.{i}.attn.qkv.bias"].chunk(3, 0)
            for j, n in enumerate(("q_proj", "k_proj", "v_proj")):
                out[f"{clip_l_prefix}encoder.layers.{i}.self_attn.{n}.weight"] = w[j]
                out[f"{clip_l_prefix}encoder.layers.{i}.self_attn.{n}.bias"] = b[j]
    if bundle.is_sdxl and getattr(bundle, "text_encoder_2", None) is not None:
        enc2 = bundle.text_encoder_2
        sd2 = enc2.state_dict()
        g_prefix = (
            XL_REFINER_G_PREFIX
            if getattr(bundle, "is_refiner", False)
            else XL_CLIP_G_PREFIX
        )
        for ldm_key, our_key in openclip_key_map(enc2).items():
            out[g_prefix + ldm_key] = sd2[our_key]
        tp = sd2.get("text_proj")
        if tp is None:
            tp = torch.eye(enc2.d_model)
        out[g_prefix + "text_projection"] = tp
    return out
