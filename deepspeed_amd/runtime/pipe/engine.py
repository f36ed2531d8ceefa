"""PipelineEngine — executes instruction schedules over a PipelineModule
(reference: deepspeed/runtime/pipe/engine.py PipelineEngine :61,
instruction dispatch _exec_schedule :1408, tied grads :275,
loss aggregation :583).

Differences from the reference, by MI355X design:
* p2p activations/grads are ISends over xGMI (direct GPU-GPU link between
  any two ranks on the node — no staging through host).
* gradient reduction composes with this framework's ZeRO-1/2 optimizer over
  the per-stage DP group; all bucket reductions are deferred to the batch
  boundary (after tied-weight all-reduce) instead of firing per-hook, so
  tied grads are correct without a separate dead-bucket path.
"""

import os

import torch

from ... import comm as dist
from ...parallel import groups
from ..engine import Engine
from . import schedule as sched
from .module import PipelineModule
from .p2p import PipeP2P


class PipelineEngine(Engine):
    def __init__(self, model: PipelineModule, **kwargs):
        assert isinstance(model, PipelineModule), \
            "PipelineEngine requires a PipelineModule"
        self.grid = model.grid
        super().__init__(model=model, mpu=self.grid, **kwargs)

        self.stage_id = self.grid.stage_id
        self.num_stages = self.grid.pipe_parallel_size
        self.micro_batches = self.gradient_accumulation_steps
        self.p2p = PipeP2P(self.device)

        self.pipe_buffers = {"inputs": [], "outputs": [], "labels": []}
        self._loss_acc = None
        self._data_iter = None
        self.loss_fn = model.loss_fn

        self._exec_map = {
            sched.LoadMicroBatch: self._exec_load_micro_batch,
            sched.ForwardPass: self._exec_forward,
            sched.BackwardPass: self._exec_backward,
            sched.SendActivation: self._exec_send_activation,
            sched.RecvActivation: self._exec_recv_activation,
            sched.SendGrad: self._exec_send_grad,
            sched.RecvGrad: self._exec_recv_grad,
            sched.ReduceTiedGrads: self._exec_reduce_tied_grads,
            sched.ReduceGrads: self._exec_reduce_grads,
            sched.OptimizerStep: self._exec_optimizer_step,
        }

    # ------------------------------------------------------------- checkpoint

    def _model_ckpt_name(self, dirname):
        """Every pipeline stage owns DIFFERENT layers, so each stage's
        dp-rank-0 writes its own stage-qualified model-states file (the
        reference writes per-layer files, pipe/module.py ckpt_layer_path;
        one file per stage is the same information at this granularity) —
        without the stage qualifier PP stages overwrite each other."""
        mp_rank = groups.get_tensor_parallel_rank()
        return os.path.join(
            dirname, f"mp_rank_{mp_rank:02d}_pp_rank_"
            f"{self.stage_id:03d}_model_states.pt")

    def _zero_ckpt_name(self, dirname):
        mp_rank = groups.get_tensor_parallel_rank()
        dp_rank = dist.get_rank(self.dp_group)
        return os.path.join(
            dirname, f"zero_pp_rank_{dp_rank}_mp_rank_{mp_rank:02d}"
            f"_stage_{self.stage_id:03d}_optim_states.pt")

    def save_checkpoint(self, save_dir, tag=None, client_state=None,
                        save_latest=True, exclude_frozen_parameters=False):
        """Stage-file save (same-PP resume) PLUS per-GLOBAL-layer module
        files (reference pipe/module.py ckpt_layer_path layer_XX files) so
        a job at a DIFFERENT pipeline degree can reassemble its stages;
        the optimizer state crosses degrees via the universal converter."""
        tag = self._ckpt_tag(tag)
        ret = super().save_checkpoint(
            save_dir, tag=tag, client_state=client_state,
            save_latest=save_latest,
            exclude_frozen_parameters=exclude_frozen_parameters)
        ckpt_dir = os.path.join(save_dir, tag)
        mp_rank = groups.get_tensor_parallel_rank()
        if dist.get_rank(self.dp_group) == 0:
            for idx, mod in self.module._layers.items():
                self.checkpoint_engine.save(
                    mod.state_dict(), self._layer_ckpt_name(
                        ckpt_dir, mp_rank, int(idx)))
            owners = self.module._tied_keys_per_stage()
            for key, mod in self.module.tied_modules.items():
                # exactly one owning stage writes (weights are synced)
                if self.stage_id == min(owners.get(key, [self.stage_id])):
                    self.checkpoint_engine.save(
                        mod.state_dict(), self._tied_ckpt_name(
                            ckpt_dir, mp_rank, key))
        if self.global_rank == 0:
            torch.save({"num_stages": self.num_stages},
                       os.path.join(ckpt_dir, "pipeline_meta.pt"))
        dist.barrier()
        return ret

    def load_checkpoint(self, load_dir, tag=None, **kwargs):
        if tag is None:
            latest = os.path.join(load_dir, "latest")
            if not os.path.exists(latest):
                return None, {}
            with open(latest) as f:
                tag = f.read().strip()
        ckpt_dir = os.path.join(load_dir, tag)
        meta_path = os.path.join(ckpt_dir, "pipeline_meta.pt")
        saved_stages = None
        if os.path.exists(meta_path):
            saved_stages = torch.load(meta_path,
                                      weights_only=False)["num_stages"]
        if saved_stages in (None, self.num_stages):
            return super().load_checkpoint(load_dir, tag=tag, **kwargs)

        # different pipeline degree: reassemble this stage's modules from
        # the per-global-layer files; optimizer state must come from the
        # universal checkpoint (native zero files are stage-shaped)
        mp_rank = groups.get_tensor_parallel_rank()
        for idx, mod in self.module._layers.items():
            sd = self.checkpoint_engine.load(
                self._layer_ckpt_name(ckpt_dir, mp_rank, int(idx)),
                map_location="cpu")
            mod.load_state_dict(sd)
        for key, mod in self.module.tied_modules.items():
            sd = self.checkpoint_engine.load(
                self._tied_ckpt_name(ckpt_dir, mp_rank, key),
                map_location="cpu")
            mod.load_state_dict(sd)
        self.module._sync_tied_weights()
        import glob as _glob
        metas = sorted(_glob.glob(os.path.join(
            ckpt_dir, f"mp_rank_{mp_rank:02d}_pp_rank_*_model_states.pt")))
        client_state = {}
        if metas:
            meta = self.checkpoint_engine.load(metas[0], map_location="cpu")
            self.global_steps = meta.get("global_steps", 0)
            self.global_samples = meta.get("global_samples", 0)
            self.skipped_steps = meta.get("skipped_steps", 0)
            client_state = meta.get("client_state", {})
        if kwargs.get("load_universal"):
            from ...checkpoint.universal import load_universal as _load_usd
            usd = _load_usd(os.path.join(load_dir, f"{tag}_universal"))
            assert hasattr(self.optimizer, "load_universal_state_dict"), \
                "cross-PP-degree optimizer resume requires ZeRO + universal"
            self.optimizer.load_universal_state_dict(self.module, usd)
        return ckpt_dir, client_state

    def _layer_ckpt_name(self, ckpt_dir, mp_rank, idx):
        return os.path.join(
            ckpt_dir, f"layer_{idx:04d}-mp_rank_{mp_rank:02d}"
            "-model_states.pt")

    def _tied_ckpt_name(self, ckpt_dir, mp_rank, key):
        return os.path.join(
            ckpt_dir, f"tied_{key}-mp_rank_{mp_rank:02d}-model_states.pt")

    # ----------------------------------------------------------------- driver

    def _reserve_buffers(self, n):
        for key in self.pipe_buffers:
            buf = self.pipe_buffers[key]
            while len(buf) < n:
                buf.append(None)

    def train_batch(self, data_iter=None):
        """Run one full batch = ``gradient_accumulation_steps`` micro-batches
        through the 1F1B schedule; returns the batch-mean loss on every rank
        (reference engine.py:train_batch:375)."""
        self.module.train()
        if data_iter is not None:
            self._data_iter = iter(data_iter) if not hasattr(
                data_iter, "__next__") else data_iter
        self._loss_acc = None
        self.set_gradient_accumulation_boundary(False)

        schedule = sched.TrainSchedule(self.micro_batches, self.num_stages,
                                       self.stage_id)
        self._reserve_buffers(schedule.num_pipe_buffers())
        for step_cmds in schedule:
            for cmd in step_cmds:
                self._exec_map[type(cmd)](cmd)
        self.p2p.flush()

        # global_steps / lr_scheduler advance inside the OptimizerStep
        # instruction (Engine.step)
        loss = self._broadcast_final_loss()
        self.global_samples += (self.train_micro_batch_size_per_gpu *
                                self.micro_batches *
                                self.grid.data_parallel_size)
        return loss

    @torch.no_grad()
    def eval_batch(self, data_iter=None):
        self.module.eval()
        if data_iter is not None:
            self._data_iter = iter(data_iter) if not hasattr(
                data_iter, "__next__") else data_iter
        self._loss_acc = None
        schedule = sched.InferenceSchedule(self.micro_batches, self.num_stages,
                                           self.stage_id)
        self._reserve_buffers(schedule.num_pipe_buffers())
        for step_cmds in schedule:
            for cmd in step_cmds:
                self._exec_map[type(cmd)](cmd)
        self.p2p.flush()
        return self._broadcast_final_loss()

    def is_first_stage(self):
        return self.stage_id == 0

    def is_last_stage(self):
        return self.stage_id == self.num_stages - 1

    def _broadcast_final_loss(self):
        """Mean micro-batch loss, broadcast from the last stage to the whole
        pipe group; averaged over the stage's DP group."""
        if self.is_last_stage():
            loss = (self._loss_acc / self.micro_batches
                    if self._loss_acc is not None
                    else torch.zeros((), device=self.device))
            loss = loss.detach().float()
            if self.grid.data_parallel_size > 1:
                dist.all_reduce(loss, group=self.grid.dp_group)
                loss = loss / self.grid.data_parallel_size
        else:
            loss = torch.zeros((), dtype=torch.float32, device=self.device)
        src = self.grid.stage_to_global(self.num_stages - 1)
        dist.broadcast(loss, src=src, group=self.grid.pp_group)
        return loss

    # ----------------------------------------------------------- instructions

    def _next_batch(self):
        assert self._data_iter is not None, "train_batch needs a data iterator"
        return next(self._data_iter)

    def _to_device(self, x):
        if torch.is_tensor(x):
            t = x.to(self.device)
            if t.is_floating_point() and self.dtype != torch.float32:
                t = t.to(self.dtype)
            return t
        if isinstance(x, (tuple, list)):
            return tuple(self._to_device(e) for e in x)
        return x

    def _exec_load_micro_batch(self, cmd):
        batch = self._next_batch()
        inputs, labels = batch if isinstance(batch, (tuple, list)) and \
            len(batch) == 2 else (batch, None)
        if self.is_first_stage():
            self.pipe_buffers["inputs"][cmd.buffer_id] = self._to_device(inputs)
        if self.is_last_stage():
            self.pipe_buffers["labels"][cmd.buffer_id] = self._to_device(labels)

    def _exec_forward(self, cmd):
        x = self.pipe_buffers["inputs"][cmd.buffer_id]
        out = self.module(x)
        if self.is_last_stage():
            labels = self.pipe_buffers["labels"][cmd.buffer_id]
            if self.loss_fn is not None and labels is not None:
                loss = self.loss_fn(out, labels)
            else:
                loss = out if torch.is_tensor(out) and out.dim() == 0 else \
                    out.float().mean()
            scaled = loss / self.micro_batches
            self.pipe_buffers["outputs"][cmd.buffer_id] = scaled
            self._loss_acc = loss.detach() + (
                self._loss_acc if self._loss_acc is not None else 0.0)
        else:
            self.pipe_buffers["outputs"][cmd.buffer_id] = out

    def _exec_backward(self, cmd):
        out = self.pipe_buffers["outputs"][cmd.buffer_id]
        if self.is_last_stage():
            if hasattr(self.optimizer, "backward"):
                self.optimizer.backward(out)
            else:
                out.backward()
        else:
            grads = self._grad_recv
            outs = (out,) if torch.is_tensor(out) else tuple(
                t for t in out if torch.is_tensor(t) and t.requires_grad)
            torch.autograd.backward(tensors=outs, grad_tensors=grads)
        self.pipe_buffers["outputs"][cmd.buffer_id] = None
        self.micro_steps += 1

    def _exec_send_activation(self, cmd):
        out = self.pipe_buffers["outputs"][cmd.buffer_id]
        tensors = (out,) if torch.is_tensor(out) else tuple(out)
        # detach for the wire; requires_grad travels in the p2p meta so the
        # receiver re-marks its buffer and backward can return input grads
        self.p2p.send(tuple(t.detach().requires_grad_(t.requires_grad)
                            for t in tensors),
                      self.grid.next_stage_rank, "act")

    def _exec_recv_activation(self, cmd):
        recvd = self.p2p.recv(self.grid.prev_stage_rank, "act")
        x = recvd[0] if len(recvd) == 1 else recvd
        self.pipe_buffers["inputs"][cmd.buffer_id] = x

    def _exec_send_grad(self, cmd):
        x = self.pipe_buffers["inputs"][cmd.buffer_id]
        tensors = (x,) if torch.is_tensor(x) else tuple(x)
        grads = tuple(t.grad for t in tensors
                      if torch.is_tensor(t) and t.grad is not None)
        assert grads, "no input grads to send — check requires_grad flow"
        self.p2p.send(grads, self.grid.prev_stage_rank, "grad")
        self.pipe_buffers["inputs"][cmd.buffer_id] = None

    def _exec_recv_grad(self, cmd):
        self._grad_recv = self.p2p.recv(self.grid.next_stage_rank, "grad")

    def _exec_reduce_tied_grads(self, cmd):
        self.module.allreduce_tied_weight_gradients()

    def _exec_reduce_grads(self, cmd):
        self.set_gradient_accumulation_boundary(True)
        if hasattr(self.optimizer, "reduce_gradients"):
            self.optimizer.reduce_gradients()
        elif self.grid.data_parallel_size > 1:
            self._buffered_allreduce_fallback()

    def _exec_optimizer_step(self, cmd):
        super(PipelineEngine, self).step()

    # the module is driven via train_batch, not forward/backward/step
    def forward(self, *a, **k):
        raise RuntimeError("PipelineEngine: use train_batch()/eval_batch()")

    def backward(self, *a, **k):
        raise RuntimeError("PipelineEngine: use train_batch()")

    def step(self, *a, **k):
        raise RuntimeError("PipelineEngine: use train_batch()")
