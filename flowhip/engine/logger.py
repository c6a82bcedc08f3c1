"""Training logger — reference-compatible files + structured metrics.

Reference behavior (train.py:102-164): per-run dir checkpoints/<name>/,
append-mode log.txt with the full args dump and a running-mean line every
SUM_FREQ=100 steps, TensorBoard scalars. This framework writes log.txt in
the same format, a metrics.jsonl stream (always — tensorboard is optional
and absent in this image), and TensorBoard events when the package exists.
Rank-0 only under DDP (callers gate).
"""

import json
import os
import time

SUM_FREQ = 100


class Logger:
    def __init__(self, scheduler, args, sum_freq=SUM_FREQ, run_dir=None):
        self.scheduler = scheduler
        self.args = vars(args) if not isinstance(args, dict) else args
        self.sum_freq = sum_freq
        self.total_steps = 0
        self.running_loss = {}
        self.writer = None
        self._t_last = time.time()
        self._imgs_since = 0

        name = self.args.get("name", "run")
        self.run_dir = run_dir or os.path.join("checkpoints", name)
        os.makedirs(self.run_dir, exist_ok=True)

        self.txt_file = open(os.path.join(self.run_dir, "log.txt"), "a")
        self.jsonl = open(os.path.join(self.run_dir, "metrics.jsonl"), "a")
        self.print_args()

    def print_args(self):
        print("\n### Experiments Arguments ###")
        self.txt_file.write("\n### Experiments Arguments ### \n")
        for k, v in self.args.items():
            print(f"{k}: {v}")
            self.txt_file.write(f"{k}: {v}\n")
        self.txt_file.flush()

    def _maybe_tb(self):
        if self.writer is None:
            try:
                from torch.utils.tensorboard import SummaryWriter
                self.writer = SummaryWriter(log_dir=self.run_dir)
            except ImportError:
                self.writer = False  # unavailable
        return self.writer or None

    def _print_training_status(self):
        keys = sorted(self.running_loss.keys())
        metrics_data = [self.running_loss[k] / self.sum_freq for k in keys]
        lr = self.scheduler.get_last_lr()[0] if self.scheduler else 0.0
        training_str = "[{:6d}, {:10.7f}] ".format(self.total_steps + 1, lr)
        metrics_str = ("{:10.4f}, " * len(metrics_data)).format(*metrics_data)

        now = time.time()
        ips = self._imgs_since / max(now - self._t_last, 1e-9)
        self._t_last, self._imgs_since = now, 0

        line = training_str + metrics_str + f" [{ips:8.2f} pairs/s]"
        print(line)
        self.txt_file.write(line + "\n")
        self.txt_file.flush()

        record = {"step": self.total_steps, "lr": lr, "pairs_per_sec": ips}
        record.update({k: self.running_loss[k] / self.sum_freq for k in keys})
        self.jsonl.write(json.dumps(record) + "\n")
        self.jsonl.flush()

        tb = self._maybe_tb()
        for k in self.running_loss:
            if tb:
                tb.add_scalar(k, self.running_loss[k] / self.sum_freq,
                              self.total_steps)
            self.running_loss[k] = 0.0

    def push(self, metrics, n_imgs=0):
        self.total_steps += 1
        self._imgs_since += n_imgs

        for key, val in metrics.items():
            self.running_loss[key] = self.running_loss.get(key, 0.0) + val

        if self.total_steps % self.sum_freq == self.sum_freq - 1:
            self._print_training_status()
            self.running_loss = {}

    def write_dict(self, results):
        tb = self._maybe_tb()
        for key, val in results.items():
            if tb:
                tb.add_scalar(key, val, self.total_steps)
            self.txt_file.write("Validation %s: %f\n" % (key, val))
        self.txt_file.flush()
        self.jsonl.write(json.dumps({"step": self.total_steps,
                                     "validation": results}) + "\n")
        self.jsonl.flush()

    def close(self):
        if self.writer:
            self.writer.close()
        self.txt_file.close()
        self.jsonl.close()
