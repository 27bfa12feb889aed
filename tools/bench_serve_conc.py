import json, os, sys, threading, time, urllib.request
sys.path.insert(0, "/root/repo")
import torch
from datatunerx_amd.serve.engine import InferenceEngine, build_model
from datatunerx_amd.serve.server import EnginePool, build_handler
from http.server import ThreadingHTTPServer

model = build_model("llama2-7b", torch.device("cuda:0"))
def run(nslots, nreq=4, toks=64):
    pool = EnginePool([InferenceEngine(model, template="llama2",
                                       device=torch.device("cuda:0"),
                                       own_stream=(nslots > 1))
                       for _ in range(nslots)])
    httpd = ThreadingHTTPServer(("127.0.0.1", 0), build_handler(pool))
    port = httpd.server_address[1]
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
    def ask(i):
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/chat/completions",
            data=json.dumps({"messages": [{"role": "user",
                                           "content": f"hello {i}"}],
                             "max_tokens": toks}).encode(),
            headers={"Content-Type": "application/json"})
        urllib.request.urlopen(req, timeout=300).read()
    ask(0)  # warm (graph capture per engine)
    if nslots > 1:
        ts = [threading.Thread(target=ask, args=(9,)) for _ in range(nslots)]
        [t.start() for t in ts]; [t.join() for t in ts]
    t0 = time.perf_counter()
    ts = [threading.Thread(target=ask, args=(i,)) for i in range(nreq)]
    [t.start() for t in ts]
    [t.join() for t in ts]
    dt = time.perf_counter() - t0
    httpd.shutdown()
    print(f"slots={nslots}: {nreq} requests x {toks} tok in {dt:.2f}s "
          f"= {nreq*toks/dt:.1f} tok/s aggregate")
run(1)
run(2)
run(4)

# batched front throughput (non-stream requests coalesce)
from datatunerx_amd.serve.server import BatchingFront
for mb in (4, 8):
    pool = EnginePool([InferenceEngine(model, template="llama2",
                                       device=torch.device("cuda:0"))])
    import datatunerx_amd.serve.engine as _e
    _e.InferenceEngine.MAX_BATCH = mb
    batcher = BatchingFront(
        InferenceEngine(model, template="llama2",
                        device=torch.device("cuda:0"), own_stream=False),
        max_batch=mb, linger=0.02)
    httpd = ThreadingHTTPServer(("127.0.0.1", 0),
                                build_handler(pool, batcher))
    port = httpd.server_address[1]
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
    def ask2(i, toks=64):
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/chat/completions",
            data=json.dumps({"messages": [{"role": "user",
                                           "content": f"hello {i}"}],
                             "max_tokens": toks}).encode(),
            headers={"Content-Type": "application/json"})
        urllib.request.urlopen(req, timeout=300).read()
    ask2(0)
    nreq = mb
    t0 = time.perf_counter()
    ts = [threading.Thread(target=ask2, args=(i,)) for i in range(nreq)]
    [t.start() for t in ts]
    [t.join() for t in ts]
    dt = time.perf_counter() - t0
    httpd.shutdown()
    print(f"batch={mb}: {nreq} requests x 64 tok in {dt:.2f}s = "
          f"{nreq*64/dt:.1f} tok/s aggregate")
