#!/usr/bin/env python3
"""Dashboard (reference parity: dashboard.py:47-2315 — Dash/Plotly app
subscribing 7+ channels into an in-memory DataStore with ~20 views).

MI355X-image stack: FastAPI + uvicorn serving (a) a JSON API exposing the
same state views (portfolio, signals, trades, risk, regime, MC,
predictions, patterns, explanations) and (b) a self-refreshing HTML
overview. Runs embedded in run_trader (same process, same bus) or
standalone against a Redis bus.

  python dashboard.py            # standalone demo on synthetic replay
"""

from __future__ import annotations

import asyncio
import json
from collections import deque

from ai_crypto_trader_amd.bus.message_bus import InProcessBus
from ai_crypto_trader_amd.bus.schema import Channels, Keys


class DataStore:
    """In-memory rolling store fed by a bus listener (reference
    dashboard.py:47-137)."""

    CHANNELS = (
        Channels.MARKET_UPDATES, Channels.SOCIAL_UPDATES,
        Channels.TRADING_SIGNALS, Channels.TRADE_EXECUTIONS,
        Channels.RISK_ALERTS, Channels.PATTERN_SIGNALS,
        Channels.NN_PREDICTIONS, Channels.EXPLAINED_TRADING_SIGNALS,
        Channels.STRATEGY_SWITCH, Channels.STRATEGY_EVOLUTION_UPDATES,
    )

    def __init__(self, bus, maxlen: int = 500, poll_s: float = 2.0):
        self.bus = bus
        self.buffers: dict[str, deque] = {
            c: deque(maxlen=maxlen) for c in self.CHANNELS
        }
        self.prices: dict[str, deque] = {}
        # per-symbol indicator series for the candlestick view
        # (close, rsi, macd, bb_position per market update)
        self.series: dict[str, dict[str, deque]] = {}
        # polled key histories for the VaR / equity views
        self.var_history: deque = deque(maxlen=1000)
        self.equity_history: deque = deque(maxlen=1000)
        self.poll_s = poll_s
        self._task = None
        self._poll_task = None

    async def start(self):
        sub = self.bus.subscribe(*self.CHANNELS)

        async def listen():
            while True:
                batch = await sub.get_batch()
                for chan, msg in batch:
                    self.buffers[chan].append(msg)
                    if chan == Channels.MARKET_UPDATES and \
                            isinstance(msg, dict) and msg.get("symbol"):
                        sym = msg["symbol"]
                        d = self.prices.setdefault(
                            sym, deque(maxlen=2000))
                        d.append(msg["current_price"])
                        s = self.series.setdefault(sym, {
                            k: deque(maxlen=600)
                            for k in ("close", "rsi", "macd",
                                      "bb_position")})
                        s["close"].append(msg.get("current_price", 0.0))
                        s["rsi"].append(msg.get("rsi", 50.0))
                        s["macd"].append(msg.get("macd", 0.0))
                        s["bb_position"].append(
                            msg.get("bb_position", 0.5))

        async def poll_keys():
            import time as _t
            while True:
                risk = await self.bus.get_json(Keys.PORTFOLIO_RISK)
                if risk:
                    self.var_history.append({
                        "t": _t.time(),
                        "var_pct": risk.get("portfolio_var_pct", 0.0),
                        "cvar_pct": risk.get("portfolio_cvar_pct",
                                             risk.get("cvar_pct", 0.0)),
                    })
                hold = await self.bus.get_json(Keys.HOLDINGS)
                if hold:
                    self.equity_history.append({
                        "t": _t.time(),
                        "total_value": hold.get("total_value", 0.0),
                    })
                await asyncio.sleep(self.poll_s)

        self._task = asyncio.create_task(listen())
        self._poll_task = asyncio.create_task(poll_keys())

    def recent(self, channel: str, n: int = 50) -> list:
        return list(self.buffers.get(channel, []))[-n:]

    def chart_data(self, symbol: str, n: int = 300) -> dict:
        """Candles synthesized from the tick series (open = previous
        close) + indicator sub-panels — the reference candlestick view's
        data contract (ref dashboard.py:509)."""
        s = self.series.get(symbol)
        if not s:
            return {"symbol": symbol, "candles": [], "rsi": [],
                    "macd": [], "bb_position": []}
        close = list(s["close"])[-n:]
        candles = []
        for i, c in enumerate(close):
            o = close[i - 1] if i else c
            candles.append({"open": o, "close": c,
                            "high": max(o, c), "low": min(o, c)})
        return {
            "symbol": symbol,
            "candles": candles,
            "rsi": list(s["rsi"])[-n:],
            "macd": list(s["macd"])[-n:],
            "bb_position": list(s["bb_position"])[-n:],
        }


VIEWS = [
    "chart", "var", "correlation", "models", "stops", "sizing",
    "explanations", "social", "equity", "mc", "regime", "patterns",
    "signals",
]
NAV = '<a href="/">home</a>' + "".join(
    f'<a href="/view/{v}">{v}</a>' for v in VIEWS)

# Self-contained canvas chart library (candlesticks, lines with guide
# levels, bar charts, heatmaps, MC fan charts) — the offline stand-in
# for the reference's Plotly dependency (dashboard.py:509-1937).
CHART_JS = r"""
async function getJSON(u){const r=await fetch(u);return r.json();}
function ctx2(id){const c=document.getElementById(id);
 const x=c.getContext('2d');x.clearRect(0,0,c.width,c.height);
 x.font='11px monospace';return [c,x];}
function span(vals){let lo=Math.min(...vals),hi=Math.max(...vals);
 if(!isFinite(lo)||!isFinite(hi)){lo=0;hi=1;}
 if(hi-lo<1e-12){hi=lo+1;}return [lo,hi];}
function drawLine(id, ys, label, fixLo, fixHi, guides){
 const [c,x]=ctx2(id); if(!ys||!ys.length){x.fillStyle='#8b949e';
  x.fillText(label+': no data yet',8,16);return;}
 let [lo,hi]= (fixLo!==undefined&&fixHi!==undefined)?[fixLo,fixHi]:span(ys);
 const W=c.width-50,H=c.height-24;
 const px=i=>40+W*i/Math.max(ys.length-1,1);
 const py=v=>8+H*(1-(v-lo)/(hi-lo));
 x.strokeStyle='#30363d';
 (guides||[]).forEach(g=>{x.beginPath();x.moveTo(40,py(g));
  x.lineTo(40+W,py(g));x.stroke();
  x.fillStyle='#8b949e';x.fillText(g,4,py(g)+3);});
 x.strokeStyle='#58a6ff';x.beginPath();
 ys.forEach((v,i)=>{i?x.lineTo(px(i),py(v)):x.moveTo(px(0),py(v));});
 x.stroke();
 x.fillStyle='#c9d1d9';x.fillText(label+'  ['+lo.toFixed(3)+' .. '+
  hi.toFixed(3)+']',8,c.height-6);}
function drawCandles(id, candles, label){
 const [c,x]=ctx2(id); if(!candles||!candles.length){
  x.fillStyle='#8b949e';x.fillText(label+': no data yet',8,16);return;}
 const los=candles.map(k=>k.low),his=candles.map(k=>k.high);
 const [lo,hi]=[Math.min(...los),Math.max(...his)];
 const W=c.width-60,H=c.height-24;
 const bw=Math.max(1,W/candles.length-1);
 const py=v=>8+H*(1-(v-lo)/((hi-lo)||1));
 candles.forEach((k,i)=>{const cx=50+W*i/candles.length;
  const up=k.close>=k.open;
  x.strokeStyle=x.fillStyle=up?'#7ee787':'#ff7b72';
  x.beginPath();x.moveTo(cx+bw/2,py(k.high));
  x.lineTo(cx+bw/2,py(k.low));x.stroke();
  const y0=py(Math.max(k.open,k.close)),y1=py(Math.min(k.open,k.close));
  x.fillRect(cx,y0,bw,Math.max(1,y1-y0));});
 x.fillStyle='#c9d1d9';x.fillText(label+'  ['+lo.toFixed(2)+' .. '+
  hi.toFixed(2)+']',8,c.height-6);}
function drawBars(id, vals, names, label){
 const [c,x]=ctx2(id); if(!vals||!vals.length){x.fillStyle='#8b949e';
  x.fillText(label+': no data yet',8,16);return;}
 const [lo,hi]=span(vals.concat([0]));
 const W=c.width-50,H=c.height-36;
 const bw=W/vals.length;
 const py=v=>8+H*(1-(v-lo)/(hi-lo));
 vals.forEach((v,i)=>{x.fillStyle=v>=0?'#58a6ff':'#ff7b72';
  const y0=py(Math.max(v,0)),y1=py(Math.min(v,0));
  x.fillRect(40+i*bw+2,y0,bw-4,Math.max(1,y1-y0));
  x.save();x.translate(40+i*bw+bw/2,c.height-4);x.rotate(-0.5);
  x.fillStyle='#8b949e';
  x.fillText(String(names[i]||i).slice(0,14),0,0);x.restore();});
 x.fillStyle='#c9d1d9';x.fillText(label,8,14);}
function drawHeatmap(id, m, names){
 const [c,x]=ctx2(id); const n=m.length; if(!n)return;
 const cs=Math.min((c.width-80)/n,(c.height-80)/n);
 for(let i=0;i<n;i++)for(let j=0;j<n;j++){
  const v=Math.max(-1,Math.min(1,m[i][j]));
  const r=v>0?Math.round(255*v):0, b=v<0?Math.round(-255*v):0;
  x.fillStyle='rgb('+r+',40,'+b+')';
  x.fillRect(70+j*cs,70+i*cs,cs-1,cs-1);
  x.fillStyle='#c9d1d9';
  if(cs>26)x.fillText(v.toFixed(2),72+j*cs,70+i*cs+cs/2);}
 x.fillStyle='#8b949e';
 names.forEach((nm,i)=>{x.fillText(String(nm).slice(0,8),2,70+i*cs+cs/2);
  x.save();x.translate(70+i*cs+cs/2,64);x.rotate(-0.6);
  x.fillText(String(nm).slice(0,8),0,0);x.restore();});}
function drawFan(id, perc, labels){
 const [c,x]=ctx2(id); const keys=Object.keys(perc);
 if(!keys.length)return;
 const all=[].concat(...keys.map(k=>perc[k]));
 const [lo,hi]=span(all); const W=c.width-60,H=c.height-24;
 const colors={'5':'#ff7b72','25':'#d29922','50':'#7ee787',
               '75':'#d29922','95':'#ff7b72'};
 keys.forEach(k=>{const ys=perc[k];
  x.strokeStyle=colors[k]||'#58a6ff';x.beginPath();
  ys.forEach((v,i)=>{const px=50+W*i/Math.max(ys.length-1,1);
   const py=8+H*(1-(v-lo)/(hi-lo));
   i?x.lineTo(px,py):x.moveTo(px,py);});x.stroke();
  x.fillStyle=colors[k]||'#58a6ff';
  x.fillText(k,c.width-36,8+H*(1-(perc[k][perc[k].length-1]-lo)/(hi-lo)));});}
function tableOf(obj){let h='<table>';
 for(const k in obj){h+='<tr><th>'+k+'</th><td>'+
  JSON.stringify(obj[k]).slice(0,300)+'</td></tr>';}
 return h+'</table>';}
function listOf(arr){let h='<table>';
 arr.slice(-15).reverse().forEach(r=>{h+='<tr><td>'+
  JSON.stringify(r).slice(0,300)+'</td></tr>';});
 return h+'</table>';}
"""


def build_app(bus, store: DataStore):
    from contextlib import asynccontextmanager

    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse

    @asynccontextmanager
    async def lifespan(_app):
        await store.start()
        yield
        for t in (store._task, store._poll_task):
            if t is not None:
                t.cancel()

    app = FastAPI(title="ai-crypto-trader-amd dashboard",
                  lifespan=lifespan)

    @app.get("/api/portfolio")
    async def portfolio():
        return {
            "holdings": await bus.get_json(Keys.HOLDINGS),
            "active_trades": await bus.get_json(Keys.ACTIVE_TRADES),
            "trailing_stops": await bus.get_json(Keys.TRAILING_STOPS),
        }

    @app.get("/api/risk")
    async def risk():
        return {
            "portfolio_risk": await bus.get_json(Keys.PORTFOLIO_RISK),
            "diversification":
                await bus.get_json(Keys.PORTFOLIO_DIVERSIFICATION),
            "adaptive_stops":
                await bus.get_json(Keys.ADAPTIVE_STOP_LOSSES),
            "alerts": store.recent(Channels.RISK_ALERTS),
        }

    @app.get("/api/signals")
    async def signals():
        return store.recent(Channels.TRADING_SIGNALS)

    @app.get("/api/trades")
    async def trades():
        return store.recent(Channels.TRADE_EXECUTIONS)

    @app.get("/api/regime")
    async def regime():
        return {
            "current": await bus.get_json(Keys.CURRENT_MARKET_REGIME),
            "history": await bus.get_json(Keys.MARKET_REGIME_HISTORY),
            "switches": store.recent(Channels.STRATEGY_SWITCH),
        }

    @app.get("/api/monte_carlo")
    async def monte_carlo():
        return {
            "results": await bus.get_json(Keys.MONTE_CARLO_RESULTS),
            "fan_chart": await bus.get_json(Keys.MC_FAN_CHART),
        }

    @app.get("/api/predictions")
    async def predictions():
        return store.recent(Channels.NN_PREDICTIONS)

    @app.get("/api/patterns")
    async def patterns():
        return {
            "report": await bus.get_json(Keys.PATTERN_ANALYSIS_REPORT),
            "signals": store.recent(Channels.PATTERN_SIGNALS),
        }

    @app.get("/api/explanations")
    async def explanations():
        return store.recent(Channels.EXPLAINED_TRADING_SIGNALS, 10)

    @app.get("/api/prices/{symbol}")
    async def prices(symbol: str):
        return list(store.prices.get(symbol, []))

    @app.get("/api/evolution")
    async def evolution():
        return {
            "updates": store.recent(Channels.STRATEGY_EVOLUTION_UPDATES),
            "params": await bus.get_json(Keys.STRATEGY_PARAMS),
        }

    @app.get("/api/social")
    async def social():
        # social panel (reference dashboard.py:759)
        metrics = await bus.hgetall(Keys.SOCIAL_METRICS)
        return {
            "metrics": {k: json.loads(v) for k, v in metrics.items()},
            "updates": store.recent(Channels.SOCIAL_UPDATES, 20),
            "risk_adjustments":
                await bus.get_json(Keys.SOCIAL_RISK_REPORT),
        }

    @app.get("/api/correlation")
    async def correlation():
        # correlation heatmap data (reference dashboard.py:1712)
        risk_d = await bus.get_json(Keys.PORTFOLIO_RISK) or {}
        return {
            "correlation_matrix": risk_d.get("correlation_matrix"),
            "symbols": risk_d.get("symbols"),
            "avg_correlation": risk_d.get("avg_correlation"),
        }

    @app.get("/api/models")
    async def models():
        # AI model performance/comparison panels (reference :1180-1479)
        return {
            "registry": await bus.get_json(Keys.MODEL_REGISTRY),
            "events": store.recent(Channels.MODEL_REGISTRY_EVENTS, 20),
            "performance":
                store.recent(Channels.MODEL_PERFORMANCE_UPDATES, 20),
        }

    # ------------------------------------------------------------------
    # Interactive views (reference dashboard.py:509-1937's ~20 Dash/
    # Plotly panels). Server-rendered HTML + a self-contained vanilla-JS
    # canvas chart library (no CDN — this stack runs offline) that
    # fetches the JSON endpoints above and redraws every few seconds.
    # ------------------------------------------------------------------

    @app.get("/api/chart/{symbol}")
    async def chart(symbol: str):
        return store.chart_data(symbol)

    @app.get("/api/var_history")
    async def var_history():
        return list(store.var_history)

    @app.get("/api/equity_history")
    async def equity_history():
        return list(store.equity_history)

    def _page(title: str, body: str, script: str) -> str:
        return f"""<html><head><title>{title}</title><style>
body{{font-family:monospace;background:#0d1117;color:#c9d1d9;margin:16px}}
canvas{{background:#161b22;border:1px solid #30363d;margin:4px 0}}
a{{color:#58a6ff;text-decoration:none;margin-right:10px}}
h2{{color:#7ee787}} table{{border-collapse:collapse;margin:8px 0}}
td,th{{border:1px solid #30363d;padding:4px 8px;font-size:13px}}
.pos{{color:#7ee787}}.neg{{color:#ff7b72}}</style></head><body>
<div>{NAV}</div><h2>{title}</h2>{body}
<script>{CHART_JS}</script><script>{script}</script></body></html>"""

    def view(path: str, title: str, body: str, script: str):
        async def handler():
            return _page(title, body, script)
        app.get(path, response_class=HTMLResponse)(handler)

    view("/view/chart", "price chart + RSI/MACD/BB (ref :509)",
         """<input id=sym value=BTCUSDC><button onclick=load()>load</button>
<canvas id=cd width=980 height=320></canvas>
<canvas id=rsi width=980 height=120></canvas>
<canvas id=macd width=980 height=120></canvas>
<canvas id=bb width=980 height=120></canvas>""",
         """async function load(){
const s=document.getElementById('sym').value;
const d=await getJSON('/api/chart/'+s);
drawCandles('cd', d.candles, s+' 1m');
drawLine('rsi', d.rsi, 'RSI', 0, 100, [30,70]);
drawLine('macd', d.macd, 'MACD');
drawLine('bb', d.bb_position, 'BB position', 0, 1, [0.05,0.95]);}
load(); setInterval(load, 5000);""")

    view("/view/var", "portfolio VaR history (ref :1485)",
         "<canvas id=v width=980 height=300></canvas>"
         "<canvas id=cv width=980 height=200></canvas>",
         """async function load(){
const h=await getJSON('/api/var_history');
drawLine('v', h.map(x=>x.var_pct), 'VaR %');
drawLine('cv', h.map(x=>x.cvar_pct), 'CVaR %');}
load(); setInterval(load, 5000);""")

    view("/view/correlation", "correlation heatmap (ref :1712)",
         "<canvas id=h width=640 height=640></canvas>",
         """async function load(){
const d=await getJSON('/api/correlation');
if(d.correlation_matrix) drawHeatmap('h', d.correlation_matrix,
                                     d.symbols||[]);}
load(); setInterval(load, 10000);""")

    view("/view/models", "AI model comparison (ref :1180-1479)",
         "<canvas id=m width=980 height=300></canvas><div id=tbl></div>",
         """async function load(){
const d=await getJSON('/api/models'); const reg=d.registry||{};
const names=Object.keys(reg.models||reg||{});
const models=reg.models||reg||{};
const scores=names.map(n=>(models[n]&&(models[n].score||
  models[n].accuracy||models[n].win_rate))||0);
drawBars('m', scores, names, 'model score');
document.getElementById('tbl').innerHTML = tableOf(models);}
load(); setInterval(load, 10000);""")

    view("/view/stops", "stop-loss visualization (ref :1592)",
         "<div id=rows></div>",
         """async function load(){
const p=await getJSON('/api/portfolio');
const at=p.active_trades||{}; const ts=p.trailing_stops||{};
let h='<table><tr><th>symbol</th><th>entry</th><th>stop</th>'+
  '<th>take-profit</th><th>trailing peak</th><th>band</th></tr>';
for(const s in at){const t=at[s]; const tr=ts[s]||{};
 const lo=t.stop_price, hi=t.tp_price, e=t.entry_price;
 const pct=x=>((x-lo)/(hi-lo)*100).toFixed(1);
 h+='<tr><td>'+s+'</td><td>'+e.toFixed(2)+'</td><td class=neg>'+
  lo.toFixed(2)+'</td><td class=pos>'+hi.toFixed(2)+'</td><td>'+
  (tr.peak||e).toFixed(2)+
  '</td><td><div style="background:#30363d;width:200px">'+
  '<div style="background:#58a6ff;height:10px;width:'+pct(e)+'%">'+
  '</div></div></td></tr>';}
document.getElementById('rows').innerHTML=h+'</table>';}
load(); setInterval(load, 3000);""")

    view("/view/sizing", "position sizing (ref :1795)",
         "<canvas id=s width=980 height=300></canvas><div id=d></div>",
         """async function load(){
const sig=await getJSON('/api/signals');
const rows=sig.filter(s=>s.risk_info).slice(-12);
drawBars('s', rows.map(s=>100*(s.risk_info.optimal_position_pct||0)),
         rows.map(s=>s.symbol), 'optimal position %');
document.getElementById('d').innerHTML=tableOf(
  Object.fromEntries(rows.map(s=>[s.symbol, s.risk_info])));}
load(); setInterval(load, 5000);""")

    view("/view/explanations", "signal explanations (ref :1937)",
         "<canvas id=w width=980 height=300></canvas><div id=x></div>",
         """async function load(){
const ex=await getJSON('/api/explanations');
if(ex.length){const e=ex[ex.length-1];
 const fw=e.factor_weights||{};
 drawBars('w', Object.values(fw), Object.keys(fw),
          (e.symbol||'')+' factor weights');
 document.getElementById('x').innerHTML='<p>'+
   (e.explanation&&e.explanation.summary||e.reasoning||'')+'</p>';}}
load(); setInterval(load, 5000);""")

    view("/view/social", "social sentiment panel (ref :759)",
         "<canvas id=sv width=980 height=260></canvas><div id=t></div>",
         """async function load(){
const d=await getJSON('/api/social');
const u=d.updates||[];
drawLine('sv', u.map(x=>(x.data&&x.data.weighted_sentiment)||0.5),
         'weighted sentiment', 0, 1, [0.5]);
document.getElementById('t').innerHTML=tableOf(d.metrics||{});}
load(); setInterval(load, 5000);""")

    view("/view/equity", "portfolio value (ref :455/:1001)",
         "<canvas id=e width=980 height=300></canvas>",
         """async function load(){
const h=await getJSON('/api/equity_history');
drawLine('e', h.map(x=>x.total_value), 'portfolio value');}
load(); setInterval(load, 5000);""")

    view("/view/mc", "Monte-Carlo fan chart (ref monte_carlo views)",
         "<canvas id=f width=980 height=360></canvas><div id=s></div>",
         """async function load(){
const d=await getJSON('/api/monte_carlo');
const fc=d.fan_chart; if(fc&&fc.percentiles){
 drawFan('f', fc.percentiles, fc.labels||[]);}
document.getElementById('s').innerHTML=tableOf(d.results||{});}
load(); setInterval(load, 10000);""")

    view("/view/regime", "market regime (ref regime views)",
         "<div id=r></div>",
         """async function load(){
const d=await getJSON('/api/regime');
let h='<p>current: <b>'+JSON.stringify(d.current)+'</b></p>';
h+='<h3>switches</h3>'+listOf(d.switches||[]);
document.getElementById('r').innerHTML=h;}
load(); setInterval(load, 5000);""")

    view("/view/patterns", "chart patterns (ref pattern views)",
         "<div id=p></div>",
         """async function load(){
const d=await getJSON('/api/patterns');
document.getElementById('p').innerHTML=
  '<h3>signals</h3>'+listOf(d.signals||[])+
  '<h3>report</h3>'+tableOf(d.report||{});}
load(); setInterval(load, 5000);""")

    view("/view/signals", "signals & trades tables (ref :880/:941)",
         "<div id=sg></div><div id=tr></div>",
         """async function load(){
const s=await getJSON('/api/signals');
const t=await getJSON('/api/trades');
document.getElementById('sg').innerHTML='<h3>signals</h3>'+listOf(s);
document.getElementById('tr').innerHTML='<h3>trades</h3>'+listOf(t);}
load(); setInterval(load, 3000);""")

    @app.get("/", response_class=HTMLResponse)
    async def index():
        holdings = await bus.get_json(Keys.HOLDINGS) or {}
        regime = await bus.get_json(Keys.CURRENT_MARKET_REGIME) or {}
        risk_d = await bus.get_json(Keys.PORTFOLIO_RISK) or {}
        sigs = store.recent(Channels.TRADING_SIGNALS, 10)
        rows = "".join(
            f"<tr><td>{s.get('symbol')}</td><td>{s.get('decision')}</td>"
            f"<td>{s.get('confidence', 0):.2f}</td></tr>"
            for s in reversed(sigs) if isinstance(s, dict))
        return f"""<html><head><title>ai-crypto-trader-amd</title>
<meta http-equiv=refresh content=5>
<style>body{{font-family:monospace;background:#0d1117;color:#c9d1d9;
margin:16px}}a{{color:#58a6ff;text-decoration:none;margin-right:10px}}
table{{border-collapse:collapse}}td,th{{border:1px solid #30363d;
padding:4px}}h2{{color:#7ee787}}</style></head><body>
<div>{NAV}</div>
<h2>ai-crypto-trader-amd</h2>
<p>portfolio: ${holdings.get('total_value', 0):,.2f} |
regime: {regime.get('regime', '?')} |
VaR: {risk_d.get('portfolio_var_pct', 0):.2f}%</p>
<h3>recent signals</h3>
<table><tr><th>symbol</th><th>decision</th><th>conf</th></tr>{rows}</table>
<p>JSON API: /api/portfolio /api/risk /api/signals /api/trades /api/regime
/api/monte_carlo /api/predictions /api/patterns /api/explanations
/api/evolution /api/chart/SYMBOL /api/var_history /api/equity_history</p>
</body></html>"""

    return app


def main():
    import uvicorn

    bus = InProcessBus()
    store = DataStore(bus)
    app = build_app(bus, store)

    async def run_all():
        # standalone demo: replay a synthetic market through the monitor
        # on the SAME event loop as the server (the in-process bus is
        # single-loop by design)
        from ai_crypto_trader_amd.config import AppConfig
        from ai_crypto_trader_amd.data.feed import SyntheticFeed
        from ai_crypto_trader_amd.data.synthetic import (
            candles_chl_v, generate_ohlcv,
        )
        from ai_crypto_trader_amd.services.market_monitor import (
            MarketMonitorService,
        )
        symbols = ["BTCUSDC", "ETHUSDC"]
        feed = SyntheticFeed(candles_chl_v(generate_ohlcv(50_000, 2)),
                             symbols, speed=60.0)
        svc = MarketMonitorService(bus, feed, AppConfig())
        await svc.start()
        server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=8050, log_level="warning"))
        await server.serve()

    asyncio.run(run_all())


if __name__ == "__main__":
    main()
