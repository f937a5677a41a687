"""Graph decode diagnostic: node count + per-replay GPU time."""
import ctypes, json, sys, time
import torch
sys.path.insert(0, "/root/repo")
from tensor_fusion_amd.models.llama import build_model, CONFIGS

model = build_model("llama3-8b", device="cuda", dtype=torch.bfloat16)
cfg = model.cfg
batch, ctx = 8, 512
total = ctx + 64
caches = model.make_kv_cache(batch, total, "cuda", torch.bfloat16)
toks = torch.randint(0, cfg.vocab, (batch, ctx), device="cuda")
model(toks, pos=torch.arange(ctx, device="cuda"), caches=caches, pos_end=ctx)
cur = torch.randint(0, cfg.vocab, (batch, 1), device="cuda")
pos_buf = torch.tensor([ctx], dtype=torch.long, device="cuda")
mask = torch.full((1, 1, 1, total), float("-inf"), device="cuda",
                  dtype=torch.bfloat16)
mask[..., :ctx] = 0.0
zero = torch.zeros(1, device="cuda", dtype=torch.bfloat16)
one = torch.ones(1, dtype=torch.long, device="cuda")

def step():
    mask.view(-1).index_copy_(0, pos_buf, zero)
    logits = model(cur, pos=pos_buf, caches=caches, mask=mask)
    cur.copy_(logits.argmax(-1))
    pos_buf.add_(one)

side = torch.cuda.Stream()
side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for _ in range(3):
        step()
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize()
try:
    g = torch.cuda.CUDAGraph(keep_graph=True)
except TypeError:
    g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
# node count via raw HIP
lib = ctypes.CDLL(None)
cnt = ctypes.c_size_t(0)
rc = -1
try:
    raw = g.raw_cuda_graph()
    rc = lib.hipGraphGetNodes(ctypes.c_void_p(raw), None, ctypes.byref(cnt))
except Exception as e:
    print("raw graph unavailable:", e)
print("nodes_rc=", rc, "node_count=", cnt.value)
try:
    hist = (ctypes.c_ulonglong * 17)()
    fn = lib.tf_graph_node_types
    if fn(ctypes.c_void_p(raw), hist) == 0:
        print("node_types:", {i: int(hist[1 + i]) for i in range(16)
                              if hist[1 + i]})
except Exception as e:
    print("no node type helper:", e)
try:
    buf = ctypes.create_string_buffer(64 << 10)
    if lib.tf_graph_kernel_histo(ctypes.c_void_p(raw), buf,
                                 ctypes.c_size_t(len(buf))) == 0:
        print("kernel histo:")
        print(buf.value.decode()[:4000])
except Exception as e:
    print("no histo helper:", e)
# replay timing
for _ in range(3):
    g.replay()
torch.cuda.synchronize()
ev = [(torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)) for _ in range(10)]
for e0, e1 in ev:
    e0.record()
    g.replay()
    e1.record()
torch.cuda.synchronize()
times = [e0.elapsed_time(e1) for e0, e1 in ev]
t0 = time.perf_counter()
for _ in range(32):
    g.replay()
torch.cuda.synchronize()
wall = (time.perf_counter() - t0) / 32 * 1000
print(json.dumps({"replay_gpu_ms": [round(t, 2) for t in times],
                  "wall_ms_per_replay": round(wall, 2)}))
