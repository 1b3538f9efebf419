import sys, torch
sys.path.insert(0, "/root/repo")
from saturn_amd.ops import require_ext
from saturn_amd.ops.functional import attention_math
ext = require_ext()

def ref_bwd(q32, k32, v32, do32, causal=True):
    q32 = q32.detach().requires_grad_(True)
    k32 = k32.detach().requires_grad_(True)
    v32 = v32.detach().requires_grad_(True)
    out = attention_math(q32, k32, v32, causal)
    out.backward(do32)
    return q32.grad, k32.grad, v32.grad

def rel(a, b):
    return ((a.float()-b.float()).norm()/b.float().norm().clamp(min=1e-9)).item()

for (B,H,T,D) in [(2,2,128,128),(2,2,128,256),(1,2,128,64)]:
    torch.manual_seed(0)
    q32 = torch.randn(B,H,T,D, device="cuda")
    k32 = torch.randn_like(q32); v32 = torch.randn_like(q32)
    do32 = torch.randn_like(q32)
    rq, rk, rv = ref_bwd(q32, k32, v32, do32)
    q = q32.to(torch.bfloat16); k = k32.to(torch.bfloat16); v = v32.to(torch.bfloat16)
    o, lse = ext.attn_fwd(q, k, v, True)
    dq, dk, dv = ext.attn_bwd(do32.to(torch.bfloat16), q, k, v, o, lse, True)
    print(f"D={D} cont: dq {rel(dq,rq):.4f} dk {rel(dk,rk):.4f} dv {rel(dv,rv):.4f}")
    # strided inputs (BTHD physical)
    qp = q.transpose(1,2).contiguous().transpose(1,2)
    kp = k.transpose(1,2).contiguous().transpose(1,2)
    vp = v.transpose(1,2).contiguous().transpose(1,2)
    o2, lse2 = ext.attn_fwd(qp, kp, vp, True)
    print(f"  fwd strided vs cont: {rel(o2, o):.6f}")
    dq2, dk2, dv2 = ext.attn_bwd(do32.to(torch.bfloat16), qp, kp, vp, o2, lse2, True)
    print(f"  bwd strided: dq {rel(dq2,rq):.4f} dk {rel(dk2,rk):.4f} dv {rel(dv2,rv):.4f}")
