import torch, sys
from uccl_amd import _load_native
C = _load_native(required=True)
g = torch.Generator().manual_seed(11)
for n in (65536, 1 << 20, 8 << 20):
    t = torch.randn(n, generator=g).to(torch.bfloat16).cuda()
    print(f"n={n} compress...", flush=True)
    frame = C.gpu_compress(t, 0)
    torch.cuda.synchronize()
    print(f"  frame {frame.numel()} bytes (ratio {t.numel()*2/frame.numel():.2f})", flush=True)
    out = torch.empty_like(t)
    C.gpu_decompress(frame, out)
    torch.cuda.synchronize()
    ok = torch.equal(t.view(torch.uint8), out.view(torch.uint8))
    print(f"  roundtrip ok={ok}", flush=True)
    if not ok:
        a = t.view(torch.uint8); b = out.view(torch.uint8)
        bad = (a != b).nonzero()
        print("  first bad:", bad[:5].flatten().tolist(), flush=True)
        sys.exit(1)
print("CODEC REPRO OK", flush=True)
