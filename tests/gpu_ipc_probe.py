"""Minimal 2-process/1-GPU HIP-IPC probe: handle exchange + peer write +
signal/poll. Isolates the IPC layer from the grpcoll protocol."""
import os, sys, socket
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
sys.path.insert(0, ".")


def worker(rank, ws, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    torch.cuda.set_device(0)
    print(f"[{rank}] init ok", flush=True)
    from magi_attention.comm.native_grpcoll import _get_handle, _open_handle
    from magi_attention import _ffa_lib
    from magi_attention._ffa_lib import check, ptr
    import ctypes

    win = torch.full((1024,), float(rank + 1), device="cuda")
    flags = torch.zeros(4, dtype=torch.int32, device="cuda")
    h = (_get_handle(win), _get_handle(flags))
    print(f"[{rank}] handles ok", flush=True)
    allh = [None] * ws
    dist.all_gather_object(allh, h, group=dist.group.WORLD)
    peer = 1 - rank
    pw = _open_handle(allh[peer][0])
    pf = _open_handle(allh[peer][1])
    print(f"[{rank}] opened peer ptrs {hex(pw)} {hex(pf)}", flush=True)

    # signal my flag, then pull peer's window after spinning on peer's flag
    check(_ffa_lib.lib().magi_grpcoll_signal(
        ptr(flags), 1, ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)), "sig")
    from magi_attention._ffa_lib import MagiGrpCollPullArgs
    pieces = torch.tensor([[0, 0, 0, 16]], dtype=torch.int32, device="cuda")
    dst = torch.zeros(16, 64, dtype=torch.float32, device="cuda")
    a = MagiGrpCollPullArgs()
    a.pieces = ptr(pieces).value
    a.n_pieces = 1
    a.row_elems = 64
    a.elem_size = 4
    a.peer_ptrs[0] = pw
    a.peer_flags[0] = pf
    a.wait_value = 1
    a.n_peers = 1
    a.dst = ptr(dst).value
    a.reduce = 0
    a.stream = torch.cuda.current_stream().cuda_stream
    check(_ffa_lib.lib().magi_grpcoll_pull(ctypes.byref(a)), "pull")
    torch.cuda.synchronize()
    expect = float(peer + 1)
    ok = bool((dst == expect).all())
    print(f"[{rank}] pull ok={ok} dst[0,0]={dst[0,0].item()}", flush=True)
    assert ok
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    s = socket.socket(); s.bind(("127.0.0.1", 0)); port = s.getsockname()[1]; s.close()
    mp.spawn(worker, args=(2, port), nprocs=2, join=True)
    print("IPC_PROBE_OK")
