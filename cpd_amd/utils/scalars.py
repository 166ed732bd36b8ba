"""Scalar logging without tensorboard (not installed in this environment;
the reference used tensorboardX, mix.py:16).  Writes TSV per tag — the same
scalar names the reference logs (loss_train/lr/loss_val/acc1/acc5) — which
tools/draw_curve.py-style postprocessing can plot anywhere."""
import os

__all__ = ["ScalarLogger"]


class ScalarLogger:
    def __init__(self, logdir):
        self.logdir = logdir
        self._files = {}
        if logdir:
            os.makedirs(logdir, exist_ok=True)

    def add_scalar(self, tag, value, step):
        if not self.logdir:
            return
        f = self._files.get(tag)
        if f is None:
            path = os.path.join(self.logdir, tag.replace("/", "_") + ".tsv")
            new = not os.path.exists(path)
            f = open(path, "a")
            if new:
                f.write("step\tvalue\n")
            self._files[tag] = f
        f.write(f"{step}\t{value}\n")
        f.flush()

    def close(self):
        for f in self._files.values():
            f.close()
        self._files = {}
