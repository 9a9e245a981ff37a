"""Runs a batch of shell commands across worker processes.

Reference analogue: utils/distribute_cli/ (run shell commands over the
distribute manager/worker layer). The MI355X deployment has no gRPC
worker pool — workers are local processes (one per GPU when
CUDA_VISIBLE_DEVICES rotation is requested), which matches how this
framework schedules everything else on one 8-GPU node.

Usage:
  python -m ydf_amd.cli.distribute_run --workers 4 --commands cmds.txt
  python -m ydf_amd.cli.distribute_run --workers 2 --gpu-rotate \\
      -- "python train_a.py" "python train_b.py"

Each line of --commands (or each positional argument) is one shell
command. Commands run concurrently on N workers; the exit code is 0
iff every command succeeded. With --gpu-rotate, worker i exports
HIP_VISIBLE_DEVICES=i%num_gpus (one command per GPU at a time).
"""
import argparse
import os
import subprocess
import sys
import threading
from queue import Queue


def run_batch(commands, workers: int = 4, gpu_rotate: bool = False,
              log=print):
    q: Queue = Queue()
    for i, c in enumerate(commands):
        q.put((i, c))
    results = [None] * len(commands)

    def count_gpus() -> int:
        try:
            import torch

            return max(torch.cuda.device_count(), 1)
        except Exception:  # noqa: BLE001
            return 1

    n_gpus = count_gpus() if gpu_rotate else 0

    def worker(slot: int):
        env = dict(os.environ)
        if gpu_rotate:
            env["HIP_VISIBLE_DEVICES"] = str(slot % n_gpus)
            env["CUDA_VISIBLE_DEVICES"] = str(slot % n_gpus)
        while True:
            try:
                i, cmd = q.get_nowait()
            except Exception:  # noqa: BLE001
                return
            r = subprocess.run(cmd, shell=True, env=env,
                               capture_output=True, text=True)
            results[i] = r.returncode
            status = "ok" if r.returncode == 0 else f"rc={r.returncode}"
            log(f"[worker {slot}] {status}: {cmd}")
            if r.returncode != 0 and r.stderr:
                log(r.stderr.strip()[-2000:])

    threads = [threading.Thread(target=worker, args=(s,))
               for s in range(max(1, workers))]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    return results


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--workers", type=int, default=4)
    ap.add_argument("--commands", help="file with one command per line")
    ap.add_argument("--gpu-rotate", action="store_true",
                    help="pin worker i to GPU i%%n via "
                         "HIP_VISIBLE_DEVICES")
    ap.add_argument("cmds", nargs="*", help="commands (alternative to "
                                            "--commands)")
    args = ap.parse_args()
    commands = list(args.cmds)
    if args.commands:
        with open(args.commands) as f:
            commands += [ln.strip() for ln in f
                         if ln.strip() and not ln.startswith("#")]
    if not commands:
        ap.error("no commands given")
    results = run_batch(commands, workers=args.workers,
                        gpu_rotate=args.gpu_rotate)
    sys.exit(0 if all(r == 0 for r in results) else 1)


if __name__ == "__main__":
    main()
