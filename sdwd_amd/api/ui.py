"""Minimal built-in web UI (ref C19, ui.py:16-404 as a single page).

The reference injected a Gradio accordion with Status / Utils / Worker
Config / Settings tabs into the host webui; this serves an equivalent
single-page control surface at ``/`` on top of the JSON API: live status
(1.5 s auto-refresh like distributed.js:7-23), a txt2img form, benchmark /
interrupt / sync-script buttons and per-worker enable/disable toggles.
"""

PAGE = """<!DOCTYPE html>
<html><head><title>sdwd_amd</title>
<style>
 body{font-family:monospace;margin:2em;background:#111;color:#ddd}
 h2{color:#e8a33d} table{border-collapse:collapse}
 td,th{border:1px solid #444;padding:4px 10px}
 button{margin:2px;background:#333;color:#ddd;border:1px solid #666;
        padding:4px 10px;cursor:pointer}
 input,select{background:#222;color:#ddd;border:1px solid #555;padding:3px}
 #log{white-space:pre-wrap;background:#000;padding:8px;max-height:220px;
      overflow-y:auto;font-size:11px}
 img{max-width:256px;margin:4px;border:1px solid #444}
</style></head>
<body>
<h2>sdwd_amd — distributed SD engine</h2>
<div id="status">loading…</div>
<h2>generate</h2>
<form onsubmit="gen(event)">
 prompt <input id="prompt" size="50" value="a herd of cows"/>
 negative <input id="negative" size="20" value=""/>
 sampler <select id="sampler"></select>
 batch <input id="batch" size="3" value="4"/>
 steps <input id="steps" size="3" value="20"/>
 size <input id="size" size="4" value="512"/>
 seed <input id="seed" size="8" value="-1"/>
 <button>generate</button>
 <button type="button" onclick="fetch('/sdapi/v1/interrupt',{method:'POST'})">
   interrupt</button>
</form>
<div id="gallery"></div>
<div>live: <img id="preview" style="max-width:128px;display:none"/></div>
<h2>model</h2>
checkpoint <select id="mainModel"></select>
vae <select id="vaeSel"></select>
<button onclick="applyModel()">apply</button> <span id="m_saved"></span>
<h2>utils</h2>
<button onclick="fetch('/sdwd/benchmark',{method:'POST'})">re-benchmark</button>
<button onclick="fetch('/sdwd/sync-script',{method:'POST'})">run sync script</button>
<h2>benchmark payload</h2>
<form onsubmit="saveBench(event)" id="benchForm">
 prompt <input id="bp_prompt" size="40"/>
 steps <input id="bp_steps" size="3"/>
 batch <input id="bp_batch_size" size="3"/>
 w <input id="bp_width" size="4"/>
 h <input id="bp_height" size="4"/>
 <button>save payload</button> <span id="bp_saved"></span>
</form>
<h2>settings</h2>
<form onsubmit="saveSettings(event)" id="settingsForm">
 job timeout <input id="s_job_timeout" size="4"/>
 <label><input type="checkbox" id="s_complement_production"/>complementary</label>
 <label><input type="checkbox" id="s_step_scaling"/>step scaling</label>
 <label><input type="checkbox" id="s_thin_client"/>thin client</label>
 <label><input type="checkbox" id="s_distribute_txt2img"/>dist txt2img</label>
 <label><input type="checkbox" id="s_distribute_img2img"/>dist img2img</label>
 <button>save</button> <span id="s_saved"></span>
</form>
<h2>log</h2><div id="log"></div>
<script>
async function refresh(){
  try{
    const s = await (await fetch('/sdwd/status')).json();
    let h = '<table><tr><th>rank</th><th>device</th><th>state</th>'+
            '<th>ipm</th><th>mpe%</th><th>pixel cap</th>'+
            '<th>model override</th><th></th></tr>';
    for(const w of s.workers){
      const mo = w.model_override || '';
      h += `<tr><td>${w.label}</td><td>${w.device}</td><td>${w.state}</td>`+
           `<td>${w.avg_ipm.toFixed(1)}</td><td>${w.mpe.toFixed(1)}</td>`+
           `<td><input id="cap_${w.label}" size="9" value="${w.pixel_cap||''}"`+
           ` placeholder="none"/></td>`+
           `<td><select id="mo_${w.label}" data-v="${mo}"></select></td>`+
           `<td><button onclick="wcfg('${w.label}')">set</button>`+
           `<button onclick="tog('${w.label}','${w.state}')">`+
           `${w.state==='DISABLED'?'enable':'disable'}</button></td></tr>`;
    }
    h += `</table><p>model: ${s.model} — vae: ${s.vae} — busy: ${s.busy}</p>`;
    document.getElementById('status').innerHTML = h;
    if (s.busy) {
      const pr = await (await fetch('/sdapi/v1/progress')).json();
      const img = document.getElementById('preview');
      if (pr.current_image) {
        img.src = 'data:image/png;base64,' + pr.current_image;
        img.style.display = 'inline';
      }
    } else {
      document.getElementById('preview').style.display = 'none';
    }
    document.getElementById('log').textContent = s.log.slice(-16).join('\\n');
  }catch(e){}
}
let MODELS = [];
function fillOverride(){
  for(const sel of document.querySelectorAll('select[id^=mo_]')){
    if (sel.options.length) continue;
    const cur = sel.dataset.v || '';
    sel.innerHTML = '<option value="">(follow main)</option>' +
      MODELS.map(m => `<option ${m===cur?'selected':''}>${m}</option>`).join('');
  }
}
async function wcfg(label){
  const cap = document.getElementById('cap_'+label).value;
  const mo = document.getElementById('mo_'+label).value;
  await fetch(`/sdwd/worker/${label}/config`,{method:'POST',
    headers:{'Content-Type':'application/json'},
    body: JSON.stringify({pixel_cap: cap ? parseInt(cap) : 0,
                          model_override: mo})});
  refresh();
}
async function loadBench(){
  try{
    const b = await (await fetch('/sdwd/benchmark-payload')).json();
    for(const k of ['prompt','steps','batch_size','width','height'])
      document.getElementById('bp_'+k).value = b[k];
  }catch(e){}
}
async function saveBench(ev){
  ev.preventDefault();
  const body = {
    prompt: document.getElementById('bp_prompt').value,
    steps: parseInt(document.getElementById('bp_steps').value),
    batch_size: parseInt(document.getElementById('bp_batch_size').value),
    width: parseInt(document.getElementById('bp_width').value),
    height: parseInt(document.getElementById('bp_height').value),
  };
  const r = await fetch('/sdwd/benchmark-payload',{method:'POST',
    headers:{'Content-Type':'application/json'}, body: JSON.stringify(body)});
  document.getElementById('bp_saved').textContent = r.ok ? 'saved' : 'error';
  setTimeout(()=>{document.getElementById('bp_saved').textContent='';}, 2000);
}
async function tog(label, state){
  const act = state==='DISABLED' ? 'enable' : 'disable';
  await fetch(`/sdwd/worker/${label}/${act}`,{method:'POST'});
  refresh();
}
async function gen(ev){
  ev.preventDefault();
  const body = {
    prompt: document.getElementById('prompt').value,
    negative_prompt: document.getElementById('negative').value,
    sampler_name: document.getElementById('sampler').value || 'Euler a',
    batch_size: parseInt(document.getElementById('batch').value),
    steps: parseInt(document.getElementById('steps').value),
    width: parseInt(document.getElementById('size').value),
    height: parseInt(document.getElementById('size').value),
    seed: parseInt(document.getElementById('seed').value),
  };
  const r = await fetch('/sdapi/v1/txt2img', {method:'POST',
    headers:{'Content-Type':'application/json'}, body: JSON.stringify(body)});
  const out = await r.json();
  document.getElementById('gallery').innerHTML =
    out.images.map(b => `<img src="data:image/png;base64,${b}"/>`).join('');
}
const BOOL_SETTINGS = ['complement_production','step_scaling','thin_client',
                       'distribute_txt2img','distribute_img2img'];
async function loadSettings(){
  try{
    const s = await (await fetch('/sdwd/settings')).json();
    document.getElementById('s_job_timeout').value = s.job_timeout;
    for(const k of BOOL_SETTINGS)
      document.getElementById('s_'+k).checked = !!s[k];
  }catch(e){}
}
async function saveSettings(ev){
  ev.preventDefault();
  const body = {job_timeout: parseFloat(
    document.getElementById('s_job_timeout').value)};
  for(const k of BOOL_SETTINGS)
    body[k] = document.getElementById('s_'+k).checked;
  const r = await fetch('/sdwd/settings',{method:'POST',
    headers:{'Content-Type':'application/json'}, body: JSON.stringify(body)});
  document.getElementById('s_saved').textContent =
    r.ok ? 'saved' : 'error';
  setTimeout(()=>{document.getElementById('s_saved').textContent='';}, 2000);
}
async function loadSamplers(){
  try{
    const ss = await (await fetch('/sdapi/v1/samplers')).json();
    const sel = document.getElementById('sampler');
    sel.innerHTML = ss.map(s =>
      `<option ${s.name==='Euler a'?'selected':''}>${s.name}</option>`
    ).join('');
  }catch(e){}
}
setInterval(refresh, 1500);  // ref distributed.js:7-23 auto-refresh cadence
async function loadModels(){
  try{
    const ms = await (await fetch('/sdapi/v1/sd-models')).json();
    MODELS = ms.map(m => m.model_name);
  }catch(e){}
}
async function loadModelSelectors(){
  try{
    const vs = await (await fetch('/sdapi/v1/sd-vae')).json();
    const opts = await (await fetch('/sdapi/v1/options')).json();
    document.getElementById('mainModel').innerHTML = MODELS.map(m =>
      `<option ${m===opts.sd_model_checkpoint?'selected':''}>${m}</option>`
    ).join('');
    document.getElementById('vaeSel').innerHTML = vs.map(v =>
      `<option ${v.model_name===opts.sd_vae?'selected':''}>`+
      `${v.model_name}</option>`).join('');
  }catch(e){}
}
async function applyModel(){
  const body = {
    sd_model_checkpoint: document.getElementById('mainModel').value,
    sd_vae: document.getElementById('vaeSel').value,
  };
  const r = await fetch('/sdapi/v1/options',{method:'POST',
    headers:{'Content-Type':'application/json'}, body: JSON.stringify(body)});
  document.getElementById('m_saved').textContent = r.ok ? 'applied' : 'error';
  setTimeout(()=>{document.getElementById('m_saved').textContent='';}, 2000);
  refresh();
}
(async () => { await loadModels(); await refresh(); fillOverride();
               loadModelSelectors(); })();
setInterval(fillOverride, 1600);
loadSettings();
loadSamplers();
loadBench();
</script>
</body></html>"""
