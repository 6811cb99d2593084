"""Static assets for the dashboard: the theme stylesheet, logo and a tiny
progressive-enhancement script (parity target: the reference's
static/style.css "Option A" theme + logo/favicon SVGs — re-designed, not
copied). Served by app.py at /static/* with long-lived cache headers; the
CSP permits only same-origin scripts/styles, so everything lives here."""

STYLE_CSS = """
:root{
  --bg:#0b0e14; --panel:#11151d; --panel2:#161b26; --line:#232a38;
  --text:#dce3ee; --dim:#8b96a8; --accent:#4f8ef7; --accent2:#8a63f4;
  --ok:#2ea06a; --warn:#d89614; --bad:#d4504c; --radius:10px;
}
*{box-sizing:border-box}
body{font-family:"Inter",system-ui,-apple-system,"Segoe UI",sans-serif;
  margin:0;background:var(--bg);color:var(--text);line-height:1.45}
a{color:var(--accent);text-decoration:none} a:hover{text-decoration:underline}
code,pre{font-family:ui-monospace,"SF Mono",Menlo,monospace;font-size:.85rem}
pre{background:var(--panel2);border:1px solid var(--line);border-radius:6px;
  padding:.7rem;overflow-x:auto;white-space:pre-wrap}

/* header */
.topbar{display:flex;align-items:center;gap:1.2rem;padding:.6rem 1.4rem;
  background:linear-gradient(90deg,#101724,#0d1320);border-bottom:1px solid var(--line);
  position:sticky;top:0;z-index:5}
.brand{display:flex;align-items:center;gap:.55rem;font-weight:700;
  font-size:1.05rem;color:var(--text)}
.brand svg{display:block}
.brand .amd{color:var(--accent);font-weight:600}
nav.main{display:flex;flex-wrap:wrap;gap:.15rem}
nav.main a{color:var(--dim);padding:.35rem .65rem;border-radius:6px;font-size:.92rem}
nav.main a:hover{color:var(--text);background:var(--panel2);text-decoration:none}
nav.main a.active{color:var(--text);background:var(--panel2)}
.topbar .spacer{flex:1}
.topbar .who{color:var(--dim);font-size:.85rem}

main{max-width:1180px;margin:0 auto;padding:1.2rem 1.4rem 3rem}
h1{font-size:1.45rem;margin:.8rem 0}
h2{font-size:1.15rem} h3{font-size:1rem;margin:.2rem 0 .6rem;color:var(--text)}
small,.dim{color:var(--dim)}

/* cards + tiles */
.card{background:var(--panel);border:1px solid var(--line);
  border-radius:var(--radius);padding:1rem 1.1rem;margin:1rem 0}
.tiles{display:grid;grid-template-columns:repeat(auto-fit,minmax(150px,1fr));
  gap:.8rem;margin:1rem 0}
.tile{background:var(--panel);border:1px solid var(--line);
  border-radius:var(--radius);padding:.8rem 1rem}
.tile .n{font-size:1.6rem;font-weight:700}
.tile .l{color:var(--dim);font-size:.8rem;text-transform:uppercase;
  letter-spacing:.06em}
.grid2{display:grid;grid-template-columns:1fr 1fr;gap:1rem}
@media(max-width:880px){.grid2{grid-template-columns:1fr}}

/* tables */
table{border-collapse:collapse;width:100%}
td,th{border-bottom:1px solid var(--line);padding:.45rem .6rem;
  text-align:left;font-size:.88rem}
th{color:var(--dim);font-weight:600;font-size:.78rem;text-transform:uppercase;
  letter-spacing:.05em}
tr:hover td{background:var(--panel2)}

/* badges */
.badge{padding:.12rem .55rem;border-radius:1rem;background:var(--panel2);
  border:1px solid var(--line);font-size:.78rem;white-space:nowrap}
.badge.block{background:#3c1514;border-color:#6b2422;color:#ff9f9b}
.badge.warn{background:#3a2c0d;border-color:#6b5312;color:#ffd778}
.badge.silent,.badge.ok{background:#11301f;border-color:#1e5b3a;color:#7fd7a4}

/* bars: span waterfall, health scores, daily chart */
.bar{height:10px;background:linear-gradient(90deg,var(--accent),var(--accent2));
  border-radius:3px;min-width:2px}
.track{background:var(--panel2);border-radius:3px;overflow:hidden}
.chart{display:flex;align-items:flex-end;gap:3px;height:90px;padding:.4rem 0}
.chart .col{flex:1;display:flex;flex-direction:column;justify-content:flex-end;
  align-items:center;gap:.2rem;min-width:8px}
.chart .col .bar{width:100%;min-height:2px}
.chart .col .lab{font-size:.6rem;color:var(--dim);transform:rotate(-45deg);
  white-space:nowrap}
.score{display:flex;align-items:center;gap:.6rem}
.score .track{flex:1;height:8px}
.score .bar.good{background:var(--ok)} .score .bar.mid{background:var(--warn)}
.score .bar.low{background:var(--bad)}

/* forms */
input,textarea,select,button{background:var(--panel2);color:var(--text);
  border:1px solid var(--line);border-radius:6px;padding:.45rem .65rem;
  margin:.2rem 0;font-size:.9rem;font-family:inherit}
input:focus,textarea:focus{outline:1px solid var(--accent)}
button{cursor:pointer;background:var(--accent);border-color:transparent;
  color:#fff;font-weight:600}
button:hover{filter:brightness(1.12)}
button.secondary{background:var(--panel2);color:var(--text);
  border-color:var(--line)}
form.inline{display:flex;gap:.5rem;align-items:center;flex-wrap:wrap}
.auth{max-width:380px;margin:8vh auto}
.auth input{width:100%}
.auth .brand{justify-content:center;margin-bottom:1rem}

footer{border-top:1px solid var(--line);color:var(--dim);font-size:.8rem;
  padding:1rem 1.4rem;text-align:center}
"""

LOGO_SVG = """<svg xmlns="http://www.w3.org/2000/svg" width="26" height="26"
viewBox="0 0 26 26" fill="none">
<rect x="1" y="1" width="24" height="24" rx="6" fill="#11151d" stroke="#4f8ef7"
 stroke-width="1.6"/>
<path d="M6 19 L6 7 L9 7 L9 12 L14 7 L18 7 L12.5 12.6 L18.5 19 L14.5 19
 L9 13.2 L9 19 Z" fill="url(#g)"/>
<defs><linearGradient id="g" x1="6" y1="7" x2="18" y2="19">
<stop stop-color="#4f8ef7"/><stop offset="1" stop-color="#8a63f4"/>
</linearGradient></defs></svg>"""

APP_JS = """
// progressive enhancement: playground runner (CSP allows same-origin only)
document.addEventListener('DOMContentLoaded', function () {
  var form = document.getElementById('pg-form');
  if (!form) return;
  form.addEventListener('submit', async function (ev) {
    ev.preventDefault();
    var out = document.getElementById('pg-out');
    out.textContent = 'running\\u2026';
    try {
      var resp = await fetch('/api/playground/run', {
        method: 'POST',
        headers: { 'Content-Type': 'application/json' },
        body: JSON.stringify({
          prompt: form.prompt.value,
          model: form.model.value || undefined,
        }),
      });
      var data = await resp.json();
      out.textContent = data.ok
        ? data.response + '\\n\\n[' + Math.round(data.latency_ms) + ' ms \\u00b7 ' +
          data.tokens_in + '/' + data.tokens_out + ' tok \\u00b7 ' +
          data.cost_usd_micro + ' \\u00b5USD \\u00b7 run ' + data.run_id + ']'
        : 'error: ' + JSON.stringify(data);
    } catch (e) { out.textContent = 'error: ' + e; }
  });
});
"""
