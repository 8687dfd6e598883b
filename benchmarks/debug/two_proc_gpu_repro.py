"""Standalone repro of the GPU-GPU snapshot hang (test_two_process_one_gpu_tcp).
Both sides print timestamped progress to stderr; hard exit after 90 s."""
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402


def log(tag, msg):
    print(f"[{time.time():.1f}] {tag}: {msg}", file=sys.stderr, flush=True)


def child(port):
    import sharedtensor_amd as st
    try:
        torch.cuda.set_device(0)
        log("child", "creating")
        seed = torch.zeros(1 << 20, device="cuda")
        t0 = time.time()
        h = st.create_or_fetch("127.0.0.1", port, seed)
        log("child", f"joined in {time.time()-t0:.1f}s")
        out = torch.zeros_like(seed)
        for i in range(20):
            h.copy_to_tensor(out)
            torch.cuda.synchronize()
            s = h.stats()
            log("child", f"iter {i} out0={out[0].item():.3f} "
                         f"recv={s['rounds_recv']} err={s['last_error']}")
            if abs(out[0].item() - 3.0) < 1e-2:
                log("child", "CONVERGED")
                break
            time.sleep(1)
        h.close()
        log("child", "closed")
    except Exception as e:
        log("child", f"EXC {e!r}")
    os._exit(0)


def main():
    import sharedtensor_amd as st
    port = 23881
    torch.cuda.set_device(0)
    log("master", "creating")
    master = st.create_or_fetch("127.0.0.1", port,
                                torch.full((1 << 20,), 3.0, device="cuda"))
    log("master", "up")
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=child, args=(port,))
    p.start()
    t0 = time.time()
    while time.time() - t0 < 90:
        s = master.stats()
        log("master", f"links={[(l['active'], l['dead'], l['rounds_sent'], l['bytes_sent']) for l in s['links']]} "
                      f"err={s['last_error']} child_alive={p.is_alive()}")
        if not p.is_alive():
            break
        time.sleep(2)
    log("master", "done; hard exit")
    os._exit(0)


if __name__ == "__main__":
    main()
