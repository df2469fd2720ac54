import sys, torch
sys.path.insert(0, ".")
from rbg_amd import ops
dev = torch.device("cuda:0")
torch.manual_seed(3)
KVH, QH, D, page = 8, 32, 128, 16
B, ctx_len, splits = 1, 16, 1
kc = torch.randn(2, KVH, page, D, dtype=torch.bfloat16, device=dev)
vc = torch.randn_like(kc)
bt = torch.tensor([[1]], dtype=torch.int32, device=dev)
ctx = torch.tensor([ctx_len], dtype=torch.int32, device=dev)

q0 = torch.zeros(B, QH, D, dtype=torch.bfloat16, device=dev)
got = ops._hip.decode_attention(q0, kc, vc, bt, ctx, 0.088, splits, 4)
want = vc[1, :, :ctx_len].float().mean(dim=1).repeat_interleave(4, dim=0)
err = (got[0].float() - want).abs()
perhead = err.max(dim=1).values
print("probeA per-head maxerr:", [round(v,3) for v in perhead.tolist()])

q1 = torch.randn(B, QH, D, dtype=torch.bfloat16, device=dev)
ref = ops._hip.decode_attention(q1, kc, vc, bt, ctx, 0.088, splits, 1)
g1 = ops._hip.decode_attention(q1, kc, vc, bt, ctx, 0.088, splits, 4)
err = (g1[0].float() - ref[0].float()).abs().max(dim=1).values
print("probeC per-head maxerr:", [round(v,3) for v in err.tolist()])
# permutation test within kvh group 0: got head h vs ref head h2
for h in range(4):
    ds = [(g1[0,h].float()-ref[0,h2].float()).abs().max().item() for h2 in range(4)]
    print(f"got h{h} vs ref h0-3:", [round(v,3) for v in ds])
