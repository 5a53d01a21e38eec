import sys, os, json, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bert_pytorch_amd.ops import extension
def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True); e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3
dev = torch.device("cuda")
for (name, K, M, N) in [("qkv",12288,3072,1024),("ffn1",12288,4096,1024),("attnout",12288,1024,1024),
                        ("qkv2",8192,3072,1024),("ffn1_2",8192,4096,1024),("attnout2",8192,1024,1024)]:
    dy = torch.randn(K, M, device=dev, dtype=torch.bfloat16)
    x = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
    for bk in (32, 64):
        os.environ["BPA_WGRAD_BK"] = str(bk)
        best = {}
        for sk in (0,1,2,3,4,6,8,12,16):
            if sk and K // sk < 256: continue
            us = timeit(lambda: extension().wgrad_tn(dy, x, sk))
            best[sk] = round(us,1)
        print(json.dumps({"shape": name, "bk": bk, "K":K, "M":M, "N":N,
                          "us_by_splitk": best}), flush=True)
