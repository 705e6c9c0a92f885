

class TestWeightCacheInvalidation:
    def test_wcat_tracks_inplace_updates(self):
        """LoRA applies with in-place add_/copy_ (pointer unchanged): the
        fused-projection concat cache must key on the tensor _version or
        q/k/v LoRA deltas are silently dropped."""
        import torch

        from sdwd_amd.models.unet import CrossAttention

        attn = CrossAttention(32, 32, 4)
        before = attn._wcat(("to_q", "to_k", "to_v")).clone()
        with torch.no_grad():
            attn.to_q.weight.add_(1.0)
        after = attn._wcat(("to_q", "to_k", "to_v"))
        assert torch.allclose(after[:32], before[:32] + 1.0)
        assert torch.equal(after[32:], before[32:])

    def test_conv_wprep_tracks_inplace_updates(self):
        import torch

        from sdwd_amd.models.layers import SDConv2d

        conv = SDConv2d(64, 64, 3, padding=1)
        a = conv._wprep().clone()
        with torch.no_grad():
            conv.weight.add_(1.0)
        b = conv._wprep()
        assert torch.allclose(b, a + 1.0)

    def test_wcat_refresh_keeps_storage(self):
        """LoRA switches must refresh the fused concat IN PLACE: captured
        hipGraphs bake the buffer's address, so a reallocation would leave
        graph replays on the previous weights."""
        import torch

        from sdwd_amd.models.unet import (
            CrossAttention, refresh_fused_projections,
        )

        attn = CrossAttention(32, 32, 4)
        first = attn._wcat(("to_q", "to_k", "to_v"))
        ptr = first.data_ptr()
        with torch.no_grad():
            attn.to_q.weight.add_(2.0)
        refresh_fused_projections(attn)
        second = attn._wcat(("to_q", "to_k", "to_v"))
        assert second.data_ptr() == ptr  # same storage (graph-visible)
        assert torch.allclose(second[:32], attn.to_q.weight)
