import multiprocessing as mp
import sys
sys.path.insert(0, "/root/repo")

def child(handle, q):
    import torch
    torch.cuda.set_device(0)
    from rayfed_amd.ops import _hip_loader
    ext = _hip_loader.load()
    ptr = ext.ipc_open(handle)
    src = ext.tensor_from_ptr(ptr, 1 << 20, 0)
    out = torch.empty(1 << 20, dtype=torch.uint8, device="cuda")
    out.copy_(src)
    torch.cuda.synchronize()
    q.put(int(out[:8].sum().item()))
    crc = ext.crc32(src)
    q.put(crc & 0xFFFFFFFF)
    ext.ipc_close(ptr)

if __name__ == "__main__":
    import torch, zlib
    torch.cuda.set_device(0)
    from rayfed_amd.ops import _hip_loader
    ext = _hip_loader.load()
    ptr, handle = ext.ipc_alloc(1 << 20)
    staging = ext.tensor_from_ptr(ptr, 1 << 20, 0)
    data = torch.arange(1 << 20, dtype=torch.float32, device="cuda").view(torch.uint8)[: 1 << 20].contiguous()
    out_crc = ext.pack_crc_async(data, staging)
    torch.cuda.synchronize()
    expect_crc = int(out_crc[2].item()) & 0xFFFFFFFF
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=child, args=(handle, q))
    p.start()
    got_sum = q.get(timeout=120)
    got_crc = q.get(timeout=60)
    p.join(30)
    print("child sum:", got_sum, "expect:", int(staging[:8].sum().item()))
    print("child crc:", hex(got_crc), "expect:", hex(expect_crc))
    assert got_crc == expect_crc
    ext.ipc_free(ptr)
    print("IPC PROBE OK")
