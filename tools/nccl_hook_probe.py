# W=1 NCCL group + overlap pipeline: exercises dist.all_reduce issued from
# autograd-hook threads over RCCL (the risky part of multi-GPU overlap)
import os, torch, torch.distributed as dist
os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29977", RANK="0",
                  WORLD_SIZE="1")
torch.cuda.set_device(0)
dist.init_process_group("nccl")
import sys; sys.path.insert(0, "/root/repo")
from cpd_amd.parallel import DistModule
from cpd_amd.trainers.core import LPTrainStep
torch.manual_seed(0)
model = torch.nn.Sequential(torch.nn.Linear(64, 128), torch.nn.ReLU(),
                            torch.nn.Linear(128, 8)).cuda()
dm = DistModule(model)
opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.1)
step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=True, overlap=3,
                   distributed=True)
crit = torch.nn.CrossEntropyLoss()
for i in range(5):
    x = torch.randn(32, 64, device="cuda")
    y = torch.randint(0, 8, (32,), device="cuda")
    step.substep(crit(dm(x), y))
torch.cuda.synchronize()
print("nccl-from-hook overlap ok")
dist.destroy_process_group()
