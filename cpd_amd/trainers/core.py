"""The customized-precision training step shared by every trainer and by
bench.py — one implementation of the reference's flagship loop (mix.py
train(), :201-358): micro-batch forward/backward into the flat bucket,
emulate-node local quantized replay, cross-rank APS low-precision
all-reduce, master-weight update.
"""
import torch
import torch.distributed as dist

from ..parallel import DistModule, sum_gradients
from ..parallel.emulate import NodeEmulator
from ..utils.master import MasterParams


class LPTrainStep:
    """Drives one optimizer step = `emulate_node` micro-batch substeps.

    Usage per micro-batch:
        done = step.substep(loss)   # loss already built from a forward pass
        # done is True on the boundary micro-batch (optimizer stepped)
    """

    def __init__(self, model: DistModule, optimizer, *, grad_exp=4, grad_man=3,
                 use_APS=True, use_kahan=False, emulate_node=1, mode="ring",
                 use_master=True, distributed=None, overlap=0):
        if model.bucket is None:
            assert emulate_node == 1, \
                "emulate_node > 1 needs the fused (fp32) bucket path"
        self.model = model
        self.optimizer = optimizer
        self.grad_exp = grad_exp
        self.grad_man = grad_man
        self.use_APS = use_APS
        self.use_kahan = use_kahan
        self.emulate_node = emulate_node
        self.mode = mode
        self.emulator = NodeEmulator(model.bucket, emulate_node) \
            if emulate_node > 1 else None
        self.master = MasterParams(optimizer) if use_master else None
        if distributed is None:
            distributed = dist.is_available() and dist.is_initialized() and \
                dist.get_world_size() > 1
        self.distributed = distributed
        self.pipeline = None
        if overlap and model.bucket is not None and emulate_node == 1:
            from ..parallel.overlap import OverlapPipeline
            self.pipeline = OverlapPipeline(
                model.bucket, grad_exp, grad_man, use_APS=use_APS,
                use_kahan=use_kahan, mode=mode, num_buckets=overlap)

    def loss_scale_denom(self):
        """The reference pre-divides the loss by world*emulate so the SUM
        reduction yields the mean (mix.py:239)."""
        world = dist.get_world_size() if self.distributed else 1
        return world * self.emulate_node

    def substep(self, loss):
        """Backward + (maybe) reduce/step.  Returns True when the optimizer
        stepped (boundary micro-batch)."""
        if self.pipeline is not None:
            self.pipeline.begin_step()
            loss.backward()
            self.pipeline.finish()  # reduction overlapped with backward
        else:
            loss.backward()
            if self.emulator is not None:
                self.emulator.store_microbatch()
                if not self.emulator.full():
                    return False
                self.emulator.reduce_(use_APS=self.use_APS,
                                      grad_exp=self.grad_exp,
                                      grad_man=self.grad_man)
            sum_gradients(self.model, use_APS=self.use_APS,
                          grad_exp=self.grad_exp, grad_man=self.grad_man,
                          use_kahan=self.use_kahan, mode=self.mode)
        if self.master is not None:
            self.master.grads_from_model()
            self.optimizer.step()
            self.master.copy_to_model()
            self.master.zero_grad()
        else:
            self.optimizer.step()
        self.model.zero_grad()
        return True
