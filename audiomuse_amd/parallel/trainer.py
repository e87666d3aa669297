"""student_clap distillation trainer (DDP over RCCL).

SURVEY.md §2.2 P8 / BASELINE config 5: the one real training path —
distill a (frozen, larger) CLAP audio teacher into the HTSAT student
with data-parallel gradient all-reduce. bf16 compute, fp32 master
params via torch DDP defaults; RCCL bucket size from config
(RCCL_BUCKET_CAP_MB — sized for per-link xGMI bandwidth, not NVSwitch).
Checkpoints: rank 0 writes epoch state.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch.nn.parallel import DistributedDataParallel as DDP

from audiomuse_amd import config as C
from audiomuse_amd.models.htsat import HTSATConfig, HTSATEncoder


@dataclass
class DistillConfig:
    batch: int = 32
    lr: float = 3e-4
    seed: int = 0
    teacher_dim_mult: int = 2       # teacher embed_dim = student * mult
    student_cfg: Optional[HTSATConfig] = None   # None = flagship config
    teacher_cfg: Optional[HTSATConfig] = None


def make_models(cfg: DistillConfig, device: torch.device):
    torch.manual_seed(cfg.seed)      # identical init on every rank
    s_cfg = cfg.student_cfg or HTSATConfig()
    student = HTSATEncoder(s_cfg).to(device)
    teacher_cfg = cfg.teacher_cfg or HTSATConfig(
        embed_dim=s_cfg.embed_dim * cfg.teacher_dim_mult,
        depths=s_cfg.depths, n_mels=s_cfg.n_mels, n_frames=s_cfg.n_frames,
        out_dim=s_cfg.out_dim)
    teacher = HTSATEncoder(teacher_cfg).to(device).eval()
    for p in teacher.parameters():
        p.requires_grad_(False)
    return student, teacher


class DistillTrainer:
    def __init__(self, cfg: Optional[DistillConfig] = None,
                 device: Optional[str] = None):
        self.cfg = cfg or DistillConfig()
        self.device = torch.device(
            device or ("cuda" if torch.cuda.is_available() else "cpu"))
        self.student, self.teacher = make_models(self.cfg, self.device)
        self.ddp = (DDP(self.student,
                        bucket_cap_mb=C.RCCL_BUCKET_CAP_MB,
                        device_ids=[self.device.index]
                        if self.device.type == "cuda" else None)
                    if dist.is_initialized() else self.student)
        self.opt = torch.optim.AdamW(self.student.parameters(),
                                     lr=self.cfg.lr)
        self.autocast_dtype = (torch.bfloat16 if self.device.type == "cuda"
                               else torch.float32)

    def synthetic_batch(self, step: int) -> torch.Tensor:
        g = torch.Generator().manual_seed(
            self.cfg.seed * 1_000_003 + step * 17 + (dist.get_rank() if
                                                     dist.is_initialized() else 0))
        s_cfg = self.student.cfg
        mel = torch.randn(self.cfg.batch, s_cfg.n_mels, s_cfg.n_frames,
                          generator=g)
        return mel.to(self.device)

    def step(self, step_idx: int) -> float:
        """One distillation step: cosine + MSE embedding matching."""
        mel = self.synthetic_batch(step_idx)
        with torch.autocast(device_type=self.device.type,
                            dtype=self.autocast_dtype,
                            enabled=self.device.type == "cuda"):
            with torch.no_grad():
                t = self.teacher(mel).float()
            s = self.ddp(mel).float()
        t = F.normalize(t, dim=1)
        s_n = F.normalize(s, dim=1)
        loss = (1.0 - (s_n * t).sum(dim=1)).mean() + 0.1 * F.mse_loss(s_n, t)
        self.opt.zero_grad(set_to_none=True)
        loss.backward()           # DDP all-reduces gradients here (RCCL)
        self.opt.step()
        return float(loss.detach())

    def save_checkpoint(self, path: str, epoch: int) -> None:
        if dist.is_initialized() and dist.get_rank() != 0:
            return
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        torch.save({"epoch": epoch,
                    "student": self.student.state_dict(),
                    "opt": self.opt.state_dict()}, path)

    def load_checkpoint(self, path: str) -> int:
        state = torch.load(path, map_location=self.device, weights_only=True)
        self.student.load_state_dict(state["student"])
        self.opt.load_state_dict(state["opt"])
        return int(state["epoch"])
