import time, torch
torch.cuda.init()
for sz_gb in [1, 4]:
    n = sz_gb << 30
    t0 = time.time(); buf = torch.empty(n, dtype=torch.uint8, pin_memory=True); t1 = time.time()
    dev = torch.empty(n, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize(); t2 = time.time()
    for _ in range(3):
        dev.copy_(buf, non_blocking=True)
    torch.cuda.synchronize(); t3 = time.time()
    print(f"{sz_gb}GB: pinned-alloc {t1-t0:.3f}s, H2D {3*sz_gb/(t3-t2):.1f} GB/s")
    # non-pinned comparison
    buf2 = torch.empty(n, dtype=torch.uint8)
    torch.cuda.synchronize(); t4 = time.time()
    dev.copy_(buf2, non_blocking=True)
    torch.cuda.synchronize(); t5 = time.time()
    print(f"{sz_gb}GB: pageable H2D {sz_gb/(t5-t4):.1f} GB/s")
    del buf, buf2, dev
