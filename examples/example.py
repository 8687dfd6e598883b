#!/usr/bin/env python3
"""Port of the reference's example workload (/root/reference/example.lua):

    Run several copies of this on one or more machines:
        python examples/example.py [host] [port]

    The first instance becomes the master; each instance adds 1.0 to every
    element of a shared 4-element tensor once per second and prints the
    replica — all replicas converge to the same (growing) values.
"""
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
import sharedtensor_amd as st  # noqa: E402


def main():
    host = sys.argv[1] if len(sys.argv) > 1 else "127.0.0.1"
    port = int(sys.argv[2]) if len(sys.argv) > 2 else 50000
    device = "cuda" if torch.cuda.is_available() else "cpu"
    seed = torch.arange(1.0, 5.0, device=device)  # torch.range(1,4) in the ref
    shared = st.create_or_fetch(host, port, seed)
    print(("master" if shared.is_master else "joined"),
          f"at {host}:{port} on {device}")
    out = torch.zeros(4, device=device)
    ones = torch.ones(4, device=device)
    try:
        while True:
            shared.copy_to_tensor(out)
            print(out.tolist(), "| staleness p50:",
                  shared.stats()["staleness_p50"])
            shared.add_from_tensor(ones)
            time.sleep(1)
    except KeyboardInterrupt:
        pass
    finally:
        shared.close()


if __name__ == "__main__":
    main()
