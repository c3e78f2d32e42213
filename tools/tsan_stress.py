"""ThreadSanitizer stress for the C++ core (run via tools/tsan_check.sh).

Hammers the classic Horovod race surface: many user
threads enqueueing collectives concurrently with the background
coordination thread and the completion callbacks — queue, handle manager,
controller, response cache, timeline.
"""
import os
import sys
import threading

sys.path.insert(0, os.environ["AXON_TSAN_REPO"])
os.environ.setdefault("AXON_RANK", "0")
os.environ.setdefault("AXON_SIZE", "1")

import torch  # noqa: E402

torch.set_num_threads(1)

import axon as hvd  # noqa: E402

hvd.init()

ERRORS = []


def worker(tid):
    try:
        for i in range(50):
            t = torch.randn(256 + tid)
            out = hvd.allreduce(t, op=hvd.Sum, name=f"t{tid}.i{i}")
            assert torch.allclose(out, t)
            h = hvd.allreduce_async_(t, average=True, name=f"t{tid}.a{i}")
            hvd.poll(h)
            hvd.synchronize(h)
    except Exception as e:  # pragma: no cover
        ERRORS.append(f"thread {tid}: {e}")


threads = [threading.Thread(target=worker, args=(k,)) for k in range(6)]
for th in threads:
    th.start()
for th in threads:
    th.join()
hvd.barrier()
hvd.shutdown()
if ERRORS:
    print("\n".join(ERRORS))
    sys.exit(1)
print("tsan stress done")
