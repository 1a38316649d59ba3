import torch, time, sys, os, itertools
sys.path.insert(0, "/root/repo")
import lws_amd.ops as ops
big = torch.empty(512*1024*1024, dtype=torch.uint8, device="cuda")
shapes = [(32,6144,4096),(32,4096,4096),(32,28672,4096),(32,4096,14336),(32,128256,4096)]
tensors = {}
for (M,N,K) in shapes:
    tensors[(M,N,K)] = (torch.randn(M,K,dtype=torch.bfloat16,device="cuda"),
                        torch.randn(N,K,dtype=torch.bfloat16,device="cuda"))
def timed(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters):
        big.fill_(0); fn()
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/iters
    t1=time.perf_counter()
    for _ in range(iters): big.fill_(0)
    torch.cuda.synchronize(); return dt-(time.perf_counter()-t1)/iters
for ksub, target in itertools.product([128,256],[512,1024,2048]):
    os.environ["LWS_SG_KSUB"]=str(ksub); os.environ["LWS_SG_TARGET"]=str(target)
    ops._SKINNY_WS.clear()
    res=[]
    for (M,N,K) in shapes:
        x,w = tensors[(M,N,K)]
        # correctness spot check
        out = ops.skinny_gemm(x,w)
        ref = (x.float()@w.float().t()).to(torch.bfloat16)
        assert torch.allclose(out.float(), ref.float(), atol=3, rtol=0.1)
        dt = timed(lambda: ops.skinny_gemm(x,w))
        res.append(f"{N}x{K}:{N*K*2/dt/1e12:.2f}")
    print(f"ksub={ksub} target={target}: " + "  ".join(res), flush=True)
