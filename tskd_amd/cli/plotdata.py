"""Live dashboard — reference bin/plotData.py rebuilt as a FastAPI web app.

The reference serves a Bokeh app on :5068 with, per patient, 10 raw plots,
10 processed plots and a prediction scatter, fed by two Spark streaming
readers and a 10-s MySQL poll (plotData.py:92-238, 301-387). Here a single
process consumes the raw channel topics and `call-stream` from the bus into
bounded in-memory series (guarded by one lock, like the reference's mutex
:58) and serves:

    GET /                  — HTML dashboard (inline JS canvas sparklines)
    GET /api/stream        — server-sent events: PUSH of raw/processed
                             snapshots every 1 s + predictions every 10 s
                             (the reference's Bokeh-websocket-push analog,
                             plotData.py:224-238); the browser subscribes
                             with EventSource and falls back to polling
                             the endpoints below if SSE drops
    GET /api/patients      — known patient ids
    GET /api/raw/<pid>     — per-channel raw series
    GET /api/processed/<pid> — per-channel processed series
    GET /api/predictions   — recent predictions from the store

Run: python -m tskd_amd.cli.plotdata [--port 5068] [--store-path ...]
"""

from __future__ import annotations

import argparse
import json
import logging
import threading
import time
from collections import defaultdict, deque

from tskd_amd.bus import Bus, Consumer
from tskd_amd.config import get_global_config
from tskd_amd.store import PredictionStore

log = logging.getLogger("plotData")

PAGE = """<!DOCTYPE html>
<html><head><title>tskd dashboard</title>
<style>body{font-family:monospace;background:#111;color:#ddd}
.ch{display:inline-block;margin:4px}canvas{background:#181c22;border:1px solid #333}
h3{margin:8px 0 2px 0}</style></head><body>
<h2>tskd ICU risk dashboard</h2><div id="root"></div>
<script>
async function j(u){return (await fetch(u)).json()}
function spark(cv,xs,color){const c=cv.getContext('2d');c.clearRect(0,0,cv.width,cv.height);
 if(!xs.length)return;const mn=Math.min(...xs),mx=Math.max(...xs),sp=(mx-mn)||1;
 c.strokeStyle=color;c.beginPath();xs.forEach((v,i)=>{const x=i/(xs.length-1||1)*cv.width,
 y=cv.height-8-(v-mn)/sp*(cv.height-16);i?c.lineTo(x,y):c.moveTo(x,y)});c.stroke();
 c.fillStyle='#888';c.fillText(mx.toFixed(1),2,10);c.fillText(mn.toFixed(1),2,cv.height-2)}
function render(p,raw,proc){const root=document.getElementById('root');
 let d=document.getElementById('p_'+p);
 if(!d){d=document.createElement('div');d.id='p_'+p;
  d.innerHTML=`<h3>${p}</h3><div class="raws"></div><div class="procs"></div>
  <div>risk: <span class="risk">-</span></div>`;root.appendChild(d)}
 for(const[name,series]of[['raws',raw],['procs',proc]]){const host=d.getElementsByClassName(name)[0];
  for(const ch in series){let w=document.getElementById(name+p+ch);
   if(!w){w=document.createElement('span');w.className='ch';w.id=name+p+ch;
    w.innerHTML=`<div>${name=='raws'?'raw':'proc'} ch${ch}</div><canvas width=160 height=60></canvas>`;
    host.appendChild(w)}
   spark(w.getElementsByTagName('canvas')[0],series[ch],name=='raws'?'#4af':'#fa4')}}}
function showPreds(ps){for(const r of ps){const d=document.getElementById('p_'+r.patient);
 if(d)d.getElementsByClassName('risk')[0].textContent=
   r.risk.toFixed(4)+' @ '+r.t.toFixed(0)+'s'}}
// PUSH path: server-sent events (1-s snapshots + 10-s predictions)
let pushOk=false;
try{const es=new EventSource('/api/stream');
 es.onmessage=(e)=>{pushOk=true;const s=JSON.parse(e.data);
  for(const p of s.patients)render(p,s.raw[p]||{},s.proc[p]||{});
  if(s.preds)showPreds(s.preds)};
 es.onerror=()=>{pushOk=false};}catch(e){pushOk=false}
// POLL fallback: only active while the push stream is down
async function tick(){if(!pushOk){const pats=await j('/api/patients');
 for(const p of pats)render(p,await j('/api/raw/'+p),await j('/api/processed/'+p))}
 setTimeout(tick,1000)}
async function preds(){if(!pushOk)showPreds(await j('/api/predictions'));
 setTimeout(preds,10000)}
setTimeout(tick,1500);setTimeout(preds,1500);
</script></body></html>"""


class DashState:
    def __init__(self, bus: Bus, cfg, store: PredictionStore,
                 starting: str = "latest", keep: int = 600):
        self.cfg = cfg
        self.store = store
        self.lock = threading.Lock()  # the reference's single mutex
        self.raw = defaultdict(lambda: defaultdict(lambda: deque(maxlen=keep)))
        self.proc = defaultdict(lambda: defaultdict(lambda: deque(maxlen=keep)))
        from tskd_amd.metrics import StageTimer
        self.pump_timer = StageTimer("plotdata.pump")
        self.raw_consumer = Consumer(bus, starting=starting)
        topics = [cfg.topic_for_channel(c) for c in cfg.channel_names]
        for t in topics:
            bus.create_topic(t)
        bus.create_topic("call-stream")
        self.raw_consumer.subscribe(topics)
        self.proc_consumer = Consumer(bus, starting=starting)
        self.proc_consumer.subscribe(["call-stream"])
        self._stop = False

    def pump(self) -> None:
        while not self._stop:
            with self.pump_timer:
                self._pump_once()

    def _pump_once(self) -> None:
        msgs = self.raw_consumer.poll(max_msgs=4096, timeout_ms=200)
        pmsgs = self.proc_consumer.poll(max_msgs=4096, timeout_ms=0)
        with self.lock:
            for m in msgs:
                try:
                    chan, val = json.loads(m.value)
                except (ValueError, TypeError):
                    continue
                self.raw[m.key.decode()][int(chan)].append(float(val))
            for m in pmsgs:
                pid, _, chan_s = m.key.decode().rpartition("_")
                try:
                    pts = json.loads(m.value)
                    self.proc[pid][int(chan_s)].extend(
                        float(p) for p in pts)
                except (ValueError, TypeError):
                    continue
        time.sleep(0.05)


def build_app(state: DashState):
    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse

    app = FastAPI(title="tskd dashboard")

    @app.get("/", response_class=HTMLResponse)
    def index():
        return PAGE

    @app.get("/health")
    def health():
        # the reference's db healthcheck analog (docker-compose.yml:138-140)
        return {"status": "ok", "predictions": state.store.count(),
                "patients": len(set(state.raw) | set(state.proc))}

    @app.get("/metrics")
    def metrics():
        # Prometheus text exposition (scrape target; reference has none)
        from fastapi.responses import PlainTextResponse
        from tskd_amd.metrics import prometheus_text
        return PlainTextResponse(prometheus_text([state.pump_timer]),
                                 media_type="text/plain; version=0.0.4")

    @app.get("/api/stream")
    async def stream(limit: int = 0):
        # PUSH: SSE snapshots — 1-s raw/processed cadence, predictions on
        # every 10th event (the reference's two Bokeh periodic callbacks,
        # plotData.py:216-219, as pushes instead of browser timers).
        # ?limit=N bounds the stream (testing/curl); 0 = endless.
        import asyncio

        from fastapi.responses import StreamingResponse

        async def gen():
            n = 0
            while limit <= 0 or n < limit:
                with state.lock:
                    pats = sorted(set(state.raw) | set(state.proc))
                    snap = {
                        "patients": pats,
                        "raw": {p: {c: list(v)[-160:]
                                    for c, v in state.raw[p].items()}
                                for p in pats},
                        "proc": {p: {c: list(v)[-160:]
                                     for c, v in state.proc[p].items()}
                                 for p in pats},
                    }
                if n % 10 == 0:
                    snap["preds"] = [
                        {"patient": p, "t": t.timestamp(), "risk": r}
                        for p, t, r in state.store.tail(100)]
                yield f"data: {json.dumps(snap)}\n\n"
                n += 1
                if limit <= 0 or n < limit:
                    await asyncio.sleep(1.0)

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.get("/api/patients")
    def patients():
        with state.lock:
            return sorted(set(state.raw) | set(state.proc))

    @app.get("/api/raw/{pid}")
    def raw(pid: str):
        with state.lock:
            return {c: list(v) for c, v in state.raw[pid].items()}

    @app.get("/api/processed/{pid}")
    def processed(pid: str):
        with state.lock:
            return {c: list(v) for c, v in state.proc[pid].items()}

    @app.get("/api/predictions")
    def predictions():
        return [{"patient": p, "t": t.timestamp(), "risk": r}
                for p, t, r in state.store.tail(100)]

    return app


def main(argv=None) -> None:
    logging.basicConfig(level=logging.INFO)
    cfg = get_global_config()
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--port", type=int, default=5068)  # reference port
    ap.add_argument("--speed", type=float, default=5.0)
    ap.add_argument("--bus-dir", default=None)
    ap.add_argument("--store-path", default="predictions.log")
    ap.add_argument("--starting", default="latest",
                    choices=["latest", "earliest"])
    args = ap.parse_args(argv)

    bus = Bus(args.bus_dir)
    store = PredictionStore(args.store_path)
    state = DashState(bus, cfg, store, starting=args.starting)
    threading.Thread(target=state.pump, daemon=True).start()
    import uvicorn
    uvicorn.run(build_app(state), host="0.0.0.0", port=args.port,
                log_level="warning")


if __name__ == "__main__":
    main()
