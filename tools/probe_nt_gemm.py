import time, json, torch
def t(fn, reps=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/reps*1e3
shapes = [(1048576,64,64),(262144,128,128),(65536,256,256),(65536,768,256),
          (16384,512,512),(65536,1024,256),(262144,512,128),(16384,2048,512)]
for M,N,K in shapes:
    dy = torch.randn(M,K,dtype=torch.bfloat16,device="cuda")
    w  = torch.randn(N,K,dtype=torch.bfloat16,device="cuda")  # dx = dy @ w.t().t()? -> w[N,K]; dx=dy@w.T? shapes: [M,K]@[K,N]
    wt = w.t().contiguous()  # [K,N]
    nt = t(lambda: torch.matmul(dy, w.t()))
    nn = t(lambda: torch.matmul(dy, wt))
    nnc= t(lambda: torch.matmul(dy, w.t().contiguous()))
    print(json.dumps({"MKN":[M,K,N],"nt_ms":round(nt,3),"nn_ms":round(nn,3),"nn_copy_ms":round(nnc,3)}))
