"""ZeRO stage 3 — parameter + gradient + optimizer-state partitioning.

Capability parity with the reference's ``deepspeed/runtime/zero/stage3.py``
(DeepSpeedZeroOptimizer_Stage3 :112), ``partition_parameters.py`` (zero.Init
/ ds-param conversion) and ``partitioned_param_coordinator.py`` (trace-based
prefetch) — re-designed MI355X-first rather than ported:

* **Partition unit = module**. Every module with direct parameters becomes a
  unit whose params live in ONE flat 16-bit buffer, padded so it splits
  evenly across the DP group. Rank r permanently stores only the slice
  ``full[r*S:(r+1)*S]`` (the *shard*); the full buffer is materialized by a
  single ``all_gather_into_tensor`` — no per-param gather lists, no
  reassembly copies. On the 8-GPU xGMI mesh this gives few, large
  collectives (a Llama-3-8B projection unit is 34–235 MB), which is what the
  7x153 GB/s point-to-point links want.
* **Gradients**: autograd accumulates into a per-unit flat grad buffer
  (param ``.grad`` are views). When the unit's last grad lands, ONE
  ``reduce_scatter_tensor`` (pre-divided by world) runs async and the
  received shard accumulates into the rank-local fp32 gradient flat — so
  gradient accumulation happens in partitioned fp32 space and the full-size
  buffers free immediately (reference: __reduce_and_partition_ipg_grads).
* **Release discipline**: full buffers free after forward and re-materialize
  just-in-time in backward via ``register_full_backward_pre_hook`` — the
  autograd-saved weight tensors are the *param objects* (leaves), so the
  ``.data`` swap is visible to backward. Releases are *sticky*: a unit is
  only freed when a different unit fetches next, so repeated calls through
  the same module (chunked lm_head loss) cost one gather.
* **Prefetch**: the first step records the unit fetch order; later steps
  prefetch the next ``prefetch_bucket_size`` elements of the trace (forward)
  / reversed trace (backward) so the all-gathers run ahead of compute on
  RCCL's internal stream.
* **Step**: identical flat-shard machinery to ZeRO-1/2 — the hand-written
  HIP fused Adam consumes the fp32 grad shard and writes the bf16 param
  shard in the same pass; there is no post-step all-gather (params gather
  lazily next forward; persistent small units refresh eagerly).
"""

import math
import os
from typing import Dict, List, Optional

import torch

from ... import accel
from ... import comm as dist
from ...utils.logging import log_dist
from ..fp16.loss_scaler import LossScalerBase, LossScaler
from ..utils import ALIGNMENT
from ...utils.nvtx import instrument_w_nvtx

FREE, INFLIGHT, AVAILABLE = 0, 1, 2


class _Unit:
    __slots__ = ("index", "name", "module", "params", "offsets", "numel",
                 "shard_size", "shard", "full", "grad_full", "grad_seen",
                 "grad_shard", "status", "handle", "persist", "group_idx",
                 "master_offset", "pending_grads", "trainable",
                 "release_pending", "in_backward", "sec_shard")

    def __init__(self, index, name, module):
        self.index = index
        self.name = name
        self.module = module
        self.params: List[torch.nn.Parameter] = []
        self.offsets: List[int] = []
        self.numel = 0
        self.shard_size = 0
        self.shard: Optional[torch.Tensor] = None
        self.full: Optional[torch.Tensor] = None
        self.grad_full: Optional[torch.Tensor] = None
        self.grad_seen: Optional[List[bool]] = None
        self.grad_shard: Optional[torch.Tensor] = None  # direct-grad mode
        self.status = FREE
        self.handle = None
        self.persist = False
        self.group_idx = 0
        self.master_offset = 0
        self.pending_grads = 0
        self.trainable = True
        self.release_pending = False
        self.in_backward = False
        self.sec_shard: Optional[torch.Tensor] = None  # hpZ secondary shard


class ZeroStage3Optimizer:
    """Module-unit sharded ZeRO-3 optimizer + parameter coordinator."""

    def __init__(self,
                 module: torch.nn.Module,
                 init_optimizer: torch.optim.Optimizer,
                 dp_group=None,
                 config=None,
                 loss_scaler: Optional[LossScalerBase] = None,
                 mpu=None):
        self.module = module
        self.optimizer = init_optimizer
        self.dp_group = dp_group
        self.world_size = dist.get_world_size(dp_group)
        self.rank = dist.get_rank(dp_group)
        # MiCS (reference zero/mics.py MiCS_Optimizer): partition params
        # over SUB-groups of mics_shard_size ranks and replicate across
        # sub-groups; gathers run inside the small shard group while
        # gradients take one extra all-reduce over the replica group.
        self.replica_group = None
        self.replica_world = 1
        mics = int(getattr(config.zero, "mics_shard_size", 0) or 0)
        if 1 <= mics < self.world_size:
            full_world = self.world_size
            grank = dist.get_rank(dp_group)
            assert full_world % mics == 0, \
                f"world {full_world} not divisible by mics_shard_size {mics}"
            for start in range(0, full_world, mics):
                ranks = list(range(start, start + mics))
                g = dist.new_group(ranks)
                if grank in ranks:
                    shard_group = g
            for off in range(mics):
                ranks = list(range(off, full_world, mics))
                g = dist.new_group(ranks)
                if grank in ranks:
                    self.replica_group = g
            self.dp_group = shard_group
            self.world_size = mics
            self.rank = grank % mics
            self.replica_world = full_world // mics
        self.mpu = mpu
        self.loss_scaler = loss_scaler or LossScaler(1.0)
        self._config_dtype = (config.dtype
                              if config.dtype != torch.float32 else None)
        zc = config.zero
        self.prefetch_bucket_size = int(zc.stage3_prefetch_bucket_size)
        self.persistence_threshold = int(zc.stage3_param_persistence_threshold)
        self.max_live_parameters = int(zc.stage3_max_live_parameters)
        self.max_reuse_distance = int(zc.stage3_max_reuse_distance)
        self.clip_grad = config.gradient_clipping
        self.overlap_comm = zc.overlap_comm
        # ZeRO-Infinity: "nvme" keeps the fp32 master + Adam moments in swap
        # files (ops/csrc/aio.cpp thread-pool engine) and streams them
        # through host RAM one chunk at a time during step(); gradients land
        # in resident host fp32 accumulators like plain CPU offload.
        self.nvme_offload = zc.offload_optimizer.device == "nvme"
        self.cpu_offload = zc.offload_optimizer.device in ("cpu", "nvme")
        self.offload_pin_memory = zc.offload_optimizer.pin_memory
        # ZeRO-Infinity parameter tier: permanent shards in pinned host
        # memory (reference offload_config OffloadParamConfig; nvme params
        # ride the same host path — the optimizer-state NVMe tier is where
        # the capacity win is, see _nvme_chunks)
        self.param_offload = zc.offload_param.device in ("cpu", "nvme")
        if self.param_offload:
            if zc.zero_quantized_weights or \
                    int(zc.zero_hpz_partition_size) > 1:
                raise ValueError("offload_param does not compose with "
                                 "qwZ/hpZ (host shards cannot feed the "
                                 "quantized/secondary gather paths)")
            if zc.offload_param.device == "nvme":
                log_dist("offload_param.device=nvme: parameter shards are "
                         "held in pinned host memory (the NVMe tier applies "
                         "to optimizer state via offload_optimizer)")
        self._swapper = None
        if self.nvme_offload:
            from ..swap_tensor.swapper import AsyncTensorSwapper
            aio = config.aio
            self._swapper = AsyncTensorSwapper(
                os.path.join(zc.offload_optimizer.nvme_path,
                             f"zero3_rank{dist.get_rank()}"),
                block_size=int(aio.block_size),
                n_threads=int(aio.thread_count))
            sub = int(zc.sub_group_size)
            self._swap_chunk = sub if sub < 10**12 else (1 << 26)
            self._nvme_init = set()   # (gi, ci) chunks whose moments exist
            self._nvme_step_count = 0
        self.is_gradient_accumulation_boundary = True
        self.overflow = False
        self.custom_loss_scaler = False
        self.micro_step_id = 0
        self._max_inflight_rs = 4
        # Without gradient accumulation the fp32 grad accumulator is pure
        # overhead: keep each unit's 16-bit reduce-scatter output and feed it
        # straight to the fused Adam kernel (which casts in-register). Saves
        # a full zero-fill + cast-add pass over the shard every step.
        self.direct_grad = (config.gradient_accumulation_steps == 1
                            and not self.cpu_offload)
        # qwZ: quantized weight all-gather (int8 + group scales over xGMI)
        self.quantized_weights = bool(zc.zero_quantized_weights) \
            and self.world_size > 1
        self.quant_group_size = int(zc.zero_quantization_group_size)
        # hpZ (ZeRO++ hierarchical partitioning, reference
        # zero/parameter_offload.py + zeropp paper): each rank additionally
        # keeps a SECONDARY shard of every unit over a small group of
        # zero_hpz_partition_size consecutive ranks (one node on multi-node
        # jobs), so forward/backward weight all-gathers traverse only the
        # intra-node xGMI links; the secondary shards refresh from the
        # primary ones with one global gather per optimizer step. Gradients
        # keep the full-world reduce-scatter (primary partitioning is
        # unchanged). Off (1) by default — on a single 8-GPU node every link
        # is xGMI so there is nothing to localize.
        # qgZ: quantized-gradient all-to-all reduce (runtime/zero/qgz.py);
        # two-level (intra-node then inter-node) when hpz groups exist
        self.quantized_grads = bool(zc.zero_quantized_gradients) \
            and self.world_size > 1
        self.qgz_inter_group = None
        self.hpz_group = None
        self.hpz_world = 1
        self.hpz_rank = 0
        hpz = int(getattr(zc, "zero_hpz_partition_size", 1) or 1)
        if hpz > 1 and self.replica_group is not None:
            log_dist("ZeRO-3: zero_hpz_partition_size ignored under MiCS "
                     "(shard groups are already node-local)")
        elif 1 < hpz < self.world_size:
            assert self.world_size % hpz == 0, \
                f"world {self.world_size} not divisible by hpz size {hpz}"
            grank = dist.get_rank(dp_group)
            for start in range(0, dist.get_world_size(dp_group), hpz):
                ranks = list(range(start, start + hpz))
                g = dist.new_group(ranks)
                if grank in ranks:
                    self.hpz_group = g
            self.hpz_world = hpz
            self.hpz_rank = self.rank % hpz
            if self.quantized_grads:
                # inter-node groups: same local index across nodes
                for off in range(hpz):
                    ranks = list(range(off, dist.get_world_size(dp_group),
                                       hpz))
                    g = dist.new_group(ranks)
                    if grank in ranks:
                        self.qgz_inter_group = g

        self.units: List[_Unit] = []
        self.param_to_unit: Dict[torch.nn.Parameter, _Unit] = {}
        self.module_to_units: Dict[torch.nn.Module, List[_Unit]] = {}
        self.group_masters: List[torch.Tensor] = []
        self.group_owned_grads: List[torch.Tensor] = []   # fp32 accumulators
        self.group_shard_numel: List[int] = []

        # trace state
        self._trace: List[int] = []          # unit indices, forward order
        self._trace_complete = False
        self._fwd_cursor = 0
        self._bwd_cursor = 0
        self._recording = True

        self._trace_misses = 0
        # fetch profiler (reference: partitioned_param_coordinator event
        # counters / ZeRO-3 fetch tracing): demand = gather launched by the
        # module's own pre-hook (prefetch arrived too late or missed),
        # prefetched = gather launched ahead by the trace walker
        self.fetch_stats = {"gathers": 0, "prefetched": 0, "demand": 0,
                            "trace_misses": 0, "steps": 0}
        self._inflight_rs = []   # (handle, recv, unit, grad_full_ref)
        self._pending_release: List[_Unit] = []
        self._hooks = []

        self._build_units()
        self._partition_all()
        self._build_masters()
        self._replace_inner_params()
        self._register_hooks()
        self.fused_adam_fn = self._try_fused_adam()

        from .partition import register_stage3
        register_stage3(self)

        # honesty about accepted-but-inert knobs (reference parity options
        # whose MI355X design doesn't need them — see module docstring)
        if int(zc.sub_group_size) != 1_000_000_000_000 and not self.nvme_offload:
            log_dist("ZeRO-3: sub_group_size is accepted but inert (module-"
                     "unit partitioning already bounds working-set size; "
                     "with offload_optimizer.device=nvme it sets the swap "
                     "chunk size)")
        if zc.round_robin_gradients:
            log_dist("ZeRO-3: round_robin_gradients is inert (per-unit "
                     "reduce-scatter has no bucket-order imbalance)")

        n_persist = sum(1 for u in self.units if u.persist)
        log_dist(f"ZeRO stage 3: world={self.world_size} units={len(self.units)} "
                 f"(persistent={n_persist}) "
                 f"shard_elems={sum(self.group_shard_numel)} "
                 f"prefetch={self.prefetch_bucket_size}"
                 + (f" hpz={self.hpz_world}" if self.hpz_group is not None
                    else ""))

    # ------------------------------------------------------------------ setup

    def _param_group_index(self, p):
        for gi, g in enumerate(self.optimizer.param_groups):
            for q in g["params"]:
                if q is p:
                    return gi
        return -1  # not in optimizer (frozen / excluded)

    def _build_units(self):
        seen = set()
        for name, mod in self.module.named_modules():
            direct = [p for p in mod._parameters.values()
                      if p is not None and id(p) not in seen]
            if not direct:
                continue
            for p in direct:
                seen.add(id(p))
            u = _Unit(len(self.units), name or "<root>", mod)
            align = ALIGNMENT * self.world_size
            off = 0
            for p in direct:
                u.params.append(p)
                u.offsets.append(off)
                off += p.numel()
            u.numel = math.ceil(off / align) * align
            u.shard_size = u.numel // self.world_size
            u.persist = u.numel <= self.persistence_threshold
            gis = {self._param_group_index(p) for p in u.params
                   if p.requires_grad}
            u.trainable = any(p.requires_grad for p in u.params)
            if len(gis) > 1:
                raise ValueError(
                    f"ZeRO-3 requires all params of module '{name}' in one "
                    f"optimizer param group (got groups {gis})")
            u.group_idx = gis.pop() if gis else -1
            self.units.append(u)
            for p in u.params:
                self.param_to_unit[p] = u
            self.module_to_units.setdefault(mod, []).append(u)

    @torch.no_grad()
    def _partition_all(self):
        """Flatten each unit, broadcast rank-0 data, keep only our shard."""
        if not self.units:
            return
        dev = accel.current_device() if accel.available() else torch.device("cpu")
        self._device = dev
        # partition in the configured compute dtype: the engine does NOT cast
        # the module for stage 3 (params may be meta/partitioned already), so
        # the cast happens here as each unit is flattened.
        dtype = self._config_dtype or self.units[0].params[0].dtype
        self._dtype = dtype
        self._empty = torch.empty(0, dtype=dtype, device=dev)
        warned_no_reset = False
        for u in self.units:
            if any(p.data.is_meta for p in u.params):
                # zero.Init(remote_device="meta"): the unit was never
                # allocated anywhere. Materialize it transiently on the
                # device, run the module's own reset_parameters (rank-0's
                # draw wins via the broadcast below), shard, free — peak
                # memory is ONE module unit, so models larger than
                # device+host memory can be constructed and trained.
                for p in u.params:
                    # set_data cannot cross meta->real; swap_tensors keeps
                    # the Parameter object identity (optimizer groups, unit
                    # lists) while materializing storage
                    new = torch.empty(p.shape, dtype=p.dtype, device=dev)
                    torch.utils.swap_tensors(p, torch.nn.Parameter(
                        new, requires_grad=p.requires_grad))
                if hasattr(u.module, "reset_parameters"):
                    u.module.reset_parameters()
                elif not warned_no_reset:
                    from ...utils.logging import logger
                    logger.warning(
                        f"meta-init module '{u.name}' has no "
                        "reset_parameters(); its weights start ZEROED — "
                        "load a checkpoint or pass an init_fn")
                    warned_no_reset = True
            full = torch.empty(u.numel, dtype=dtype, device=dev)
            for p, off in zip(u.params, u.offsets):
                full[off:off + p.numel()].copy_(p.data.reshape(-1).to(dev, dtype))
            tail = u.offsets[-1] + u.params[-1].numel()
            if tail < u.numel:
                full[tail:].zero_()
            if self.replica_group is not None:
                # MiCS: replicas must start identical too
                dist.broadcast(full, src=0)
            elif self.world_size > 1:
                dist.broadcast(full,
                               src=dist.get_global_rank(self.dp_group, 0),
                               group=self.dp_group)
            if self.world_size > 1:
                u.shard = full[self.rank * u.shard_size:
                               (self.rank + 1) * u.shard_size].clone()
                if self.hpz_group is not None and not u.persist:
                    ss = u.numel // self.hpz_world
                    u.sec_shard = full[self.hpz_rank * ss:
                                       (self.hpz_rank + 1) * ss].clone()
            else:
                # ws=1: the shard IS the full buffer — keep it, no clone
                u.shard = full
            if self.param_offload and not u.persist:
                # ZeRO-Infinity parameter tier: the permanent shard lives in
                # pinned host memory; gathers stage it H2D first. Persistent
                # (small) units stay device-resident — offloading them
                # would cost a host round-trip per step for ~no memory.
                host = torch.empty(u.shard.shape, dtype=u.shard.dtype,
                                   device="cpu")
                if self.offload_pin_memory and accel.available():
                    host = host.pin_memory()
                host.copy_(u.shard)
                u.shard = host
            for p in u.params:
                p.ds_shape = p.shape
                p.ds_numel = p.numel()
                p.data = self._empty
            u.full = None
            u.status = FREE
            del full
        if accel.available():
            torch.cuda.empty_cache()
        # safe mode: unit count/sizes must agree before any collective
        from ...utils.safe_mode import checked
        checked([u.numel for u in self.units], group=self.dp_group,
                what="ZeRO-3 unit sizes")
        # persistent units stay materialized
        for u in self.units:
            if u.persist:
                self._launch_gather(u)
                self._make_available(u)

    def _build_masters(self):
        ngroups = len(self.optimizer.param_groups)
        totals = [0] * ngroups
        for u in self.units:
            if u.group_idx < 0 or not u.trainable:
                continue
            u.master_offset = totals[u.group_idx]
            totals[u.group_idx] += u.shard_size
        master_dev = torch.device("cpu") if self.cpu_offload else self._device
        for gi in range(ngroups):
            n_resident = 0 if self.nvme_offload else totals[gi]
            m = torch.empty(n_resident, dtype=torch.float32, device=master_dev)
            if self.direct_grad:
                g = torch.empty(0, dtype=torch.float32, device=master_dev)
            else:
                g = torch.zeros(totals[gi], dtype=torch.float32,
                                device=master_dev)
            if self.cpu_offload and self.offload_pin_memory and accel.available():
                if m.numel():
                    m = m.pin_memory()
                g = g.pin_memory()
            self.group_masters.append(m)
            self.group_owned_grads.append(g)
            self.group_shard_numel.append(totals[gi])
        if self.nvme_offload:
            self._nvme_write_initial_masters()
            return
        for u in self.units:
            if u.group_idx < 0 or not u.trainable:
                continue
            dst = self.group_masters[u.group_idx][
                u.master_offset:u.master_offset + u.shard_size]
            dst.copy_(u.shard.float() if not self.cpu_offload
                      else u.shard.float().cpu())

    def _replace_inner_params(self):
        for gi, group in enumerate(self.optimizer.param_groups):
            master = self.group_masters[gi]
            if master.numel() == 0:
                group["params"] = []
            else:
                mp = torch.nn.Parameter(master, requires_grad=False)
                group["params"] = [mp]
                self.group_masters[gi] = mp

    def _try_fused_adam(self):
        try:
            from ...ops.adam import multi_tensor_adam_available, fused_adam_step
            if self.cpu_offload:
                # offload step runs on host: needs the AVX cpu_adam op
                from ...ops._loader import get_ext
                ext = get_ext()
                if ext is not None and hasattr(ext, "cpu_adam_flat"):
                    return fused_adam_step
                return None
            if multi_tensor_adam_available():
                return fused_adam_step
        except Exception:
            pass
        return None

    # ------------------------------------------------------------ fetch/release

    def _unit_quant_group(self, numel: int) -> int:
        g = self.quant_group_size
        while g > 2 and numel % g:
            g //= 2
        return g

    @instrument_w_nvtx
    def _launch_gather(self, u: _Unit):
        if u.status != FREE:
            return
        self.fetch_stats["gathers"] += 1
        # hpZ: gather from the node-local secondary shard over the small
        # group instead of the primary shard over the whole world
        if u.sec_shard is not None:
            src, group, gworld = u.sec_shard, self.hpz_group, self.hpz_world
        else:
            src, group, gworld = u.shard, self.dp_group, self.world_size
        if src.device.type == "cpu" and self._device.type != "cpu":
            # offload_param: stage the pinned host shard H2D; stream order
            # sequences the copy before the allgather / first use
            src = src.to(self._device, non_blocking=True)
        if self.world_size > 1 and self.quantized_weights:
            from ...ops.quantizer import quantize
            gs = self._unit_quant_group(src.numel())
            q, s = quantize(src, gs, bits=8)
            q_full = torch.empty(q.numel() * gworld, dtype=q.dtype,
                                 device=q.device)
            s_full = torch.empty(s.numel() * gworld,
                                 dtype=torch.float32, device=q.device)
            h1 = dist.all_gather_into_tensor(q_full, q, group=group,
                                             async_op=True)
            h2 = dist.all_gather_into_tensor(s_full, s, group=group,
                                             async_op=True)
            u.full = None
            u.handle = ("qwz", h1, h2, q_full, s_full, gs)
        elif self.world_size > 1:
            u.full = torch.empty(u.numel, dtype=self._dtype,
                                 device=self._device)
            u.handle = dist.all_gather_into_tensor(u.full, src,
                                                   group=group,
                                                   async_op=True)
        else:
            # ws=1: shard covers the whole unit — alias when device-resident
            # (no alloc, no copy); with offload_param, src is the fresh H2D
            # staging copy
            u.full = src
            u.handle = None
        u.status = INFLIGHT

    def _make_available(self, u: _Unit):
        if u.status == AVAILABLE:
            return
        assert u.status == INFLIGHT, f"unit {u.name} not in flight"
        if isinstance(u.handle, tuple):  # qwZ quantized gather
            from ...ops.quantizer import dequantize
            _, h1, h2, q_full, s_full, gs = u.handle
            if h1 is not None:
                h1.wait()
            if h2 is not None:
                h2.wait()
            u.full = dequantize(q_full, s_full, u.numel, gs, bits=8,
                                dtype=self._dtype)
            u.handle = None
        elif u.handle is not None:
            u.handle.wait()
            u.handle = None
        for p, off in zip(u.params, u.offsets):
            p.data = u.full[off:off + p.ds_numel].view(p.ds_shape)
        u.status = AVAILABLE

    def _release(self, u: _Unit):
        if u.persist or u.status == FREE:
            return
        for p in u.params:
            p.data = self._empty
        u.full = None
        u.handle = None
        u.status = FREE
        u.release_pending = False

    def _flush_pending_releases(self, keep=(), phase=None):
        if not self._pending_release:
            return
        still = []
        for u in self._pending_release:
            if u in keep or u.in_backward:
                still.append(u)
            elif u.release_pending:
                if phase is not None and self._reuse_within(u, phase):
                    still.append(u)   # next use is close — keep resident
                else:
                    self._release(u)
        self._pending_release = still

    def _reuse_within(self, u, phase):
        """stage3_max_reuse_distance (reference partitioned_param_coordinator
        __params_to_release): skip releasing a unit whose next use in the
        recorded trace is within the distance budget (elements) — tied/
        chunk-reused modules would otherwise re-gather immediately."""
        if not self._trace_complete or self.max_reuse_distance <= 0:
            return False
        t, c = (self._trace, self._fwd_cursor) if phase == "fwd" else \
            (self._rtrace, self._bwd_cursor)
        elems = 0
        for i in range(c, len(t)):
            if t[i] == u.index:
                return elems < self.max_reuse_distance
            elems += self.units[t[i]].numel
            if elems >= self.max_reuse_distance:
                return False
        return False

    # ------------------------------------------------------------------ hooks

    def _register_hooks(self):
        for mod, units in self.module_to_units.items():
            h1 = mod.register_forward_pre_hook(self._pre_forward_hook)
            h2 = mod.register_forward_hook(self._post_forward_hook)
            h3 = mod.register_full_backward_pre_hook(self._pre_backward_hook)
            self._hooks += [h1, h2, h3]
        for u in self.units:
            for i, (p, off) in enumerate(zip(u.params, u.offsets)):
                if p.requires_grad:
                    self._hooks.append(p.register_post_accumulate_grad_hook(
                        self._make_grad_hook(u, i, off)))
            u.pending_grads = sum(1 for p in u.params if p.requires_grad)

    def _units_for(self, mod):
        units = list(self.module_to_units.get(mod, ()))
        # shared/tied params owned by another module's unit
        for p in mod._parameters.values():
            if p is not None:
                u = self.param_to_unit.get(p)
                if u is not None and u not in units:
                    units.append(u)
        return units

    def _pre_forward_hook(self, mod, inputs):
        units = self._units_for(mod)
        for u in units:
            u.release_pending = False
            if u.status == FREE:
                self.fetch_stats["demand"] += 1
            self._launch_gather(u)
        self._flush_pending_releases(keep=units, phase="fwd")
        if self._recording:
            for u in units:
                self._trace.append(u.index)
        else:
            self._advance_fwd_cursor(units)
            self._prefetch(self._trace, self._fwd_cursor)
        for u in units:
            self._make_available(u)
        self._drain_inflight_rs(limit=self._max_inflight_rs)

    def _post_forward_hook(self, mod, inputs, output):
        for u in self._units_for(mod):
            if not u.persist:
                u.release_pending = True
                if u not in self._pending_release:
                    self._pending_release.append(u)

    def _pre_backward_hook(self, mod, grad_output):
        if not torch.is_grad_enabled() and not torch.is_inference_mode_enabled():
            pass
        units = self._units_for(mod)
        for u in units:
            u.release_pending = False
            u.in_backward = True
            if u.status == FREE:
                self.fetch_stats["demand"] += 1
            self._launch_gather(u)
        self._flush_pending_releases(keep=units, phase="bwd")
        if self._trace_complete:
            self._advance_bwd_cursor(units)
            self._prefetch(self._rtrace, self._bwd_cursor)
        for u in units:
            self._make_available(u)
            self._ensure_grad_buffer(u)
        self._drain_inflight_rs(limit=self._max_inflight_rs)

    def _ensure_grad_buffer(self, u: _Unit):
        """Allocate the unit's flat grad landing buffer UNZEROED. Grads are
        copied in by the post-accumulate hook (first arrival = copy, later
        arrivals = add), which removes the per-microstep fill kernel and the
        autograd `p.grad +=` read-modify-write of the preset-view scheme."""
        if not u.trainable:
            return
        if u.grad_full is None:
            u.grad_full = torch.empty(u.numel, dtype=self._dtype,
                                      device=self._device)
            u.grad_seen = [False] * len(u.params)

    def _make_grad_hook(self, u: _Unit, idx: int, off: int):
        def hook(param):
            if u.grad_full is None:
                self._ensure_grad_buffer(u)
            if param.grad is not None:
                g = param.grad.detach().reshape(-1)
                dst = u.grad_full[off:off + param.ds_numel]
                if u.grad_seen[idx]:
                    dst.add_(g)
                else:
                    dst.copy_(g)
                    u.grad_seen[idx] = True
                param.grad = None
            u.pending_grads -= 1
            if u.pending_grads == 0:
                self._reduce_unit(u)
        return hook

    # -------------------------------------------------------------- reduction

    @instrument_w_nvtx
    def _reduce_unit(self, u: _Unit):
        """reduce-scatter one unit's grads; accumulate shard into fp32."""
        if u.grad_full is None:
            return
        grad = u.grad_full
        # landing buffer was left unzeroed: clear spans that never received a
        # grad this microstep (frozen/unused params) and the alignment tail
        for (p, off, seen) in zip(u.params, u.offsets, u.grad_seen):
            if not seen:
                grad[off:off + p.ds_numel].zero_()
        tail = u.offsets[-1] + u.params[-1].ds_numel
        if tail < u.numel:
            grad[tail:].zero_()
        if self.world_size > 1 and self.quantized_grads:
            from .qgz import quantized_reduce
            grad.div_(self.world_size)
            recv = quantized_reduce(grad, u.shard_size, self.dp_group,
                                    intra_group=self.hpz_group,
                                    inter_group=self.qgz_inter_group,
                                    group_size=self.quant_group_size)
            h = None
        elif self.world_size > 1:
            grad.div_(self.world_size)
            recv = torch.empty(u.shard_size, dtype=grad.dtype,
                               device=grad.device)
            h = dist.reduce_scatter_tensor(recv, grad, group=self.dp_group,
                                           async_op=True)
        else:
            recv = grad
            h = None
        self._inflight_rs.append((h, recv, u, grad))
        # params not needed anymore this micro-step
        u.grad_full = None
        u.grad_seen = None
        for p in u.params:
            p.grad = None
        u.in_backward = False
        u.pending_grads = sum(1 for p in u.params if p.requires_grad)
        if not u.persist:
            self._release(u)

    def _drain_inflight_rs(self, limit=0):
        while len(self._inflight_rs) > limit:
            h, recv, u, grad_ref = self._inflight_rs.pop(0)
            if h is not None:
                h.wait()
            if u.group_idx >= 0:
                if self.replica_group is not None:
                    # MiCS phase 2: average this shard across replicas
                    dist.all_reduce(recv, group=self.replica_group)
                    recv = recv / self.replica_world if recv.dtype.is_floating_point else recv
                if self.direct_grad:
                    # keep the 16-bit RS output; Adam consumes it directly
                    u.grad_shard = recv
                    continue
                dst = self.group_owned_grads[u.group_idx][
                    u.master_offset:u.master_offset + u.shard_size]
                if self.cpu_offload:
                    dst.add_(recv.float().cpu())
                else:
                    # mixed-dtype add_: the cast fuses into the add kernel
                    # instead of materializing a separate fp32 temp
                    dst.add_(recv)

    # -------------------------------------------------------------- prefetch

    def _advance_fwd_cursor(self, units):
        t = self._trace
        for u in units:
            i = self._fwd_cursor
            while i < len(t) and t[i] != u.index:
                i += 1
            if i < len(t):
                # skipped entries = module calls the trace predicted but the
                # model did not make (dynamic control flow)
                self._trace_misses += i - self._fwd_cursor
                self._fwd_cursor = i + 1
            else:
                self._trace_misses += 1  # unit not in trace at all

    def _advance_bwd_cursor(self, units):
        t = self._rtrace
        for u in units:
            i = self._bwd_cursor
            while i < len(t) and t[i] != u.index:
                i += 1
            if i < len(t):
                self._bwd_cursor = i + 1

    @instrument_w_nvtx
    def _prefetch(self, trace, cursor):
        budget = self.prefetch_bucket_size
        i = cursor
        while i < len(trace) and budget > 0:
            u = self.units[trace[i]]
            if u.status == FREE:
                self.fetch_stats["prefetched"] += 1
                self._launch_gather(u)
                budget -= u.numel
            i += 1

    def _end_step_trace(self):
        if self._recording and self._trace:
            self._trace_complete = True
            self._recording = False
            # dedup consecutive repeats (chunked lm_head)
            dedup = []
            for idx in self._trace:
                if not dedup or dedup[-1] != idx:
                    dedup.append(idx)
            self._trace = dedup
            self._rtrace = list(reversed(dedup))
            # safe mode: a divergent fetch order across ranks means the
            # next step's coalesced gathers would deadlock — assert now
            from ...utils.safe_mode import checked
            checked(self._trace, group=self.dp_group, what="ZeRO-3 fetch trace")
        elif self._trace_complete and self._trace_misses > \
                max(4, len(self._trace) // 4):
            # trace invalidation (reference partitioned_param_coordinator
            # trace-mismatch handling): the model's call order diverged from
            # the recorded trace — prefetch was fetching the wrong units.
            # Correctness is unaffected (pre-forward hooks always gather on
            # demand) but memory/bandwidth churns, so re-record next step.
            log_dist(f"ZeRO-3: prefetch trace invalidated "
                     f"({self._trace_misses} misses) — re-recording")
            self._trace = []
            self._rtrace = []
            self._trace_complete = False
            self._recording = True
        self.fetch_stats["trace_misses"] += self._trace_misses
        self.fetch_stats["steps"] += 1
        if os.environ.get("DS_AMD_FETCH_PROFILE") == "1":
            s = self.fetch_stats
            log_dist(f"ZeRO-3 fetch: gathers={s['gathers']} "
                     f"prefetched={s['prefetched']} demand={s['demand']} "
                     f"trace_misses={s['trace_misses']} steps={s['steps']}")
        self._trace_misses = 0
        self._fwd_cursor = 0
        self._bwd_cursor = 0

    # ------------------------------------------------------------------- api

    def backward(self, loss, retain_graph=False):
        self.micro_step_id += 1
        if self.custom_loss_scaler:
            (loss * self.external_loss_scale).backward(retain_graph=retain_graph)
        else:
            self.loss_scaler.backward(loss.float(), retain_graph=retain_graph)
        self._post_backward()

    def _post_backward(self):
        # flush units whose grads partially arrived (unused params)
        for u in self.units:
            if u.grad_full is not None and u.in_backward:
                self._reduce_unit(u)
        self._drain_inflight_rs(limit=0)
        self._flush_pending_releases()
        self._end_step_trace()

    def reduce_gradients(self):
        self._drain_inflight_rs(limit=0)

    @torch.no_grad()
    @instrument_w_nvtx
    def step(self, closure=None):
        assert closure is None, "closure not supported"
        self._drain_inflight_rs(limit=0)

        scale = self.loss_scaler.loss_scale
        if self.direct_grad:
            owned = [u.grad_shard for u in self.units
                     if u.grad_shard is not None and u.grad_shard.numel() > 0]
        else:
            owned = [g for g in self.group_owned_grads if g.numel() > 0]
        norm_sq = None
        if owned:
            norms = torch._foreach_norm(owned, 2.0)
            norm_sq = torch.stack([n.float() for n in norms]).pow(2).sum()
            if dist.is_initialized() and self.world_size > 1:
                if norm_sq.is_cuda or not self.cpu_offload:
                    dist.all_reduce(norm_sq, group=self.dp_group)
                else:
                    t = norm_sq.to(self._device)
                    dist.all_reduce(t, group=self.dp_group)
                    norm_sq = t.cpu()
            if self.mpu is not None:
                dist.all_reduce(norm_sq, group=self.mpu.get_model_parallel_group())
        self.overflow = bool(norm_sq is not None and
                             (torch.isinf(norm_sq) or torch.isnan(norm_sq)))
        self.loss_scaler.update_scale(self.overflow)
        if self.overflow:
            log_dist(f"overflow detected, skipping step "
                     f"(new loss scale {self.loss_scaler.loss_scale})")
            self._zero_owned_grads()
            self.micro_step_id = 0
            return

        global_norm = (norm_sq.sqrt() / scale) if norm_sq is not None else None
        combined_scale = scale
        if self.clip_grad > 0 and global_norm is not None:
            clip = (global_norm / self.clip_grad).clamp(min=1.0)
            combined_scale = scale * clip
        self._global_grad_norm = float(global_norm) if global_norm is not None else 0.0

        stepped = wrote_params = False
        if self.nvme_offload:
            self._nvme_step(combined_scale)
            stepped = wrote_params = True
        elif self.fused_adam_fn is not None:
            stepped, wrote_params = self._fused_step(combined_scale)
        if not stepped:
            self._torch_step(combined_scale)
        if not wrote_params:
            self._copy_masters_to_shards()
        self._refresh_persistent()
        self._refresh_secondary()
        self._zero_owned_grads()
        self.micro_step_id = 0

    def _fused_step(self, combined_scale):
        wrote_params = True
        for gi, group in enumerate(self.optimizer.param_groups):
            master = self.group_masters[gi]
            if master.numel() == 0:
                continue
            segments = []
            for u in self.units:
                if u.group_idx != gi or not u.trainable:
                    continue
                out16 = u.shard if u.shard.dtype == torch.bfloat16 else None
                if out16 is not None and out16.device != master.device:
                    # param-only offload (host shard, device master): the
                    # fused kernel can't write across; fall back to the
                    # post-step D2H copy
                    out16 = None
                if out16 is None:
                    wrote_params = False
                if self.direct_grad:
                    g16 = u.grad_shard
                    if g16 is None:  # unit never produced grads this step
                        g16 = torch.zeros(u.shard_size, dtype=self._dtype,
                                          device=self._device)
                    segments.append((u.master_offset, u.shard_size, out16, g16))
                else:
                    segments.append((u.master_offset, u.shard_size, out16))
            ok = self.fused_adam_fn(self.optimizer, group, master,
                                    self.group_owned_grads[gi],
                                    combined_scale, segments=segments)
            if not ok:
                return False, False
        return True, wrote_params

    # ------------------------------------------------ NVMe (ZeRO-Infinity)

    def _nvme_chunks(self, gi):
        total = self.group_shard_numel[gi]
        ci = c0 = 0
        while c0 < total:
            yield ci, c0, min(c0 + self._swap_chunk, total)
            ci += 1
            c0 += self._swap_chunk

    def _units_in_span(self, gi, c0, c1):
        for u in self.units:
            if u.group_idx != gi or not u.trainable:
                continue
            a = max(u.master_offset, c0)
            z = min(u.master_offset + u.shard_size, c1)
            if a < z:
                yield u, a, z

    @torch.no_grad()
    def _nvme_write_initial_masters(self):
        for gi in range(len(self.group_shard_numel)):
            for ci, c0, c1 in self._nvme_chunks(gi):
                chunk = torch.empty(c1 - c0, dtype=torch.float32)
                for u, a, z in self._units_in_span(gi, c0, c1):
                    src = u.shard[a - u.master_offset:z - u.master_offset]
                    chunk[a - c0:z - c0].copy_(src.float().cpu())
                self._swapper.swap_out(f"g{gi}c{ci}_p", chunk)
                self._swapper.synchronize()
                del chunk

    @torch.no_grad()
    def _nvme_step(self, combined_scale):
        """Chunked optimizer step over swapped state (reference
        swap_tensor/optimizer_utils.py OptimizerSwapper): for each chunk of
        the group flat, read master+moments from NVMe, run the AVX host Adam
        against the resident fp32 grad slice, write the updated bf16 params
        straight into the overlapping unit shards, and write master+moments
        back out. Host RAM high-water mark is one chunk (x3 fp32 + 1 bf16),
        independent of model size."""
        from ...ops.adam import _adam_hyperparams, _torch_adam_step
        from ...ops._loader import get_ext
        ext = get_ext()
        have_cpu_adam = ext is not None and hasattr(ext, "cpu_adam_flat")
        scale = float(combined_scale) if not torch.is_tensor(combined_scale) \
            else float(combined_scale.item())
        inv_scale = 1.0 / scale
        self._nvme_step_count += 1
        step = self._nvme_step_count
        for gi, group in enumerate(self.optimizer.param_groups):
            if self.group_shard_numel[gi] == 0:
                continue
            lr, beta1, beta2, eps, wd, adamw = _adam_hyperparams(
                self.optimizer, group)
            grads = self.group_owned_grads[gi]
            for ci, c0, c1 in self._nvme_chunks(gi):
                n = c1 - c0
                p = self._swapper.swap_in(f"g{gi}c{ci}_p")
                if (gi, ci) in self._nvme_init:
                    m = self._swapper.swap_in(f"g{gi}c{ci}_m")
                    v = self._swapper.swap_in(f"g{gi}c{ci}_v")
                    self._swapper.synchronize()
                else:
                    self._swapper.synchronize()
                    m = torch.zeros(n, dtype=torch.float32)
                    v = torch.zeros(n, dtype=torch.float32)
                    self._nvme_init.add((gi, ci))
                g = grads[c0:c1]
                w16 = torch.empty(n, dtype=torch.bfloat16)
                if have_cpu_adam:
                    ext.cpu_adam_flat(p, g, m, v, w16, lr, beta1, beta2,
                                      eps, wd, step, inv_scale, adamw)
                else:
                    _torch_adam_step(p, g, m, v, lr, beta1, beta2, eps,
                                     wd, step, adamw, inv_scale)
                    w16.copy_(p)
                for u, a, z in self._units_in_span(gi, c0, c1):
                    dst = u.shard[a - u.master_offset:z - u.master_offset]
                    dst.copy_(w16[a - c0:z - c0], non_blocking=dst.is_cuda)
                self._swapper.swap_out(f"g{gi}c{ci}_p", p)
                self._swapper.swap_out(f"g{gi}c{ci}_m", m)
                self._swapper.swap_out(f"g{gi}c{ci}_v", v)
                self._swapper.synchronize()
                del p, m, v, w16
        if accel.available():
            torch.cuda.synchronize()

    @torch.no_grad()
    def _nvme_flat(self, gi, kind):
        """Materialize one group's fp32 flat (param/exp_avg/exp_avg_sq) from
        swap — checkpointing only."""
        out = torch.empty(self.group_shard_numel[gi], dtype=torch.float32)
        for ci, c0, c1 in self._nvme_chunks(gi):
            key = f"g{gi}c{ci}_{kind}"
            if kind != "p" and (gi, ci) not in self._nvme_init:
                out[c0:c1].zero_()
                continue
            self._swapper.swap_in(key, out[c0:c1])
            self._swapper.synchronize()
        return out

    @torch.no_grad()
    def _nvme_load_flat(self, gi, kind, flat):
        for ci, c0, c1 in self._nvme_chunks(gi):
            self._swapper.swap_out(f"g{gi}c{ci}_{kind}", flat[c0:c1].clone())
            self._swapper.synchronize()
            if kind != "p":
                self._nvme_init.add((gi, ci))
        if kind == "p":  # refresh bf16 unit shards from the new master
            for u, a, z in self._units_in_span(gi, 0,
                                               self.group_shard_numel[gi]):
                dst = u.shard[a - u.master_offset:z - u.master_offset]
                dst.copy_(flat[a:z].to(dst.dtype))

    def _torch_step(self, combined_scale):
        from ...ops.adam import _torch_adam_step  # noqa
        if self.direct_grad:
            self._materialize_owned_grads()
        for gi, group in enumerate(self.optimizer.param_groups):
            master = self.group_masters[gi]
            if master.numel() == 0:
                continue
            g = self.group_owned_grads[gi]
            if isinstance(combined_scale, torch.Tensor) or combined_scale != 1.0:
                g = g / combined_scale
            master.grad = g
        self.optimizer.step()
        for gi in range(len(self.optimizer.param_groups)):
            if self.group_masters[gi].numel() > 0:
                self.group_masters[gi].grad = None

    def _copy_masters_to_shards(self):
        for u in self.units:
            if u.group_idx < 0 or not u.trainable:
                continue
            master = self.group_masters[u.group_idx]
            src = master.data[u.master_offset:u.master_offset + u.shard_size]
            u.shard.copy_(src, non_blocking=self.cpu_offload)

    def _refresh_secondary(self):
        """hpZ: re-derive the node-local secondary shards from the updated
        primary shards — ONE global all-gather per trainable unit per step,
        after which every fetch this step (fwd + bwd + grad-accum
        microsteps) is intra-node only."""
        if self.hpz_group is None:
            return
        ss_div = self.hpz_world
        for u in self.units:
            if u.sec_shard is None or not u.trainable:
                continue
            full = torch.empty(u.numel, dtype=self._dtype, device=self._device)
            dist.all_gather_into_tensor(full, u.shard, group=self.dp_group)
            ss = u.numel // ss_div
            u.sec_shard.copy_(full[self.hpz_rank * ss:(self.hpz_rank + 1) * ss])
            del full

    def _refresh_persistent(self):
        handles = []
        for u in self.units:
            if u.persist and u.status == AVAILABLE and u.trainable:
                if self.world_size > 1:
                    handles.append(dist.all_gather_into_tensor(
                        u.full, u.shard, group=self.dp_group, async_op=True))
                # ws=1: full aliases shard — the step updated it in place
        for h in handles:
            if h is not None:
                h.wait()

    def _zero_owned_grads(self):
        if self.direct_grad:
            for u in self.units:
                u.grad_shard = None
        for g in self.group_owned_grads:
            if g.numel():
                g.zero_()

    def _materialize_owned_grads(self):
        """direct-grad fallback for the unfused torch step: scatter the
        per-unit 16-bit shards into fp32 group accumulators."""
        for gi in range(len(self.group_owned_grads)):
            if self.group_owned_grads[gi].numel() != self.group_shard_numel[gi]:
                self.group_owned_grads[gi] = torch.zeros(
                    self.group_shard_numel[gi], dtype=torch.float32,
                    device=self._device if not self.cpu_offload
                    else torch.device("cpu"))
            else:
                self.group_owned_grads[gi].zero_()
        for u in self.units:
            if u.group_idx < 0 or u.grad_shard is None:
                continue
            dst = self.group_owned_grads[u.group_idx][
                u.master_offset:u.master_offset + u.shard_size]
            dst.copy_(u.grad_shard)

    def zero_grad(self, set_to_none: bool = False):
        pass  # transient grad buffers; fp32 accumulators zeroed in step()

    # --------------------------------------------------------------- plumbing

    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @property
    def loss_scale(self):
        return self.loss_scaler.loss_scale

    @property
    def state(self):
        return self.optimizer.state

    def get_global_grad_norm(self):
        return getattr(self, "_global_grad_norm", 0.0)

    # ---------------------------------------------------- gather context/API

    class _Gathered:
        """GatheredParameters equivalent (reference partition_parameters.py
        :2121): materialize units covering `params` inside the context; write
        modifications back to the shards on exit when modifier_rank says so."""

        def __init__(self, opt, params, modifier_rank=None):
            self.opt = opt
            if isinstance(params, torch.nn.Parameter):
                params = [params]
            self.units = []
            for p in params:
                u = opt.param_to_unit.get(p)
                if u is not None and u not in self.units:
                    self.units.append(u)
            self.modifier_rank = modifier_rank

        def __enter__(self):
            # pin as persistent for the duration so the forward-hook release
            # machinery cannot free the buffers mid-context (e.g. when a
            # hybrid-engine generate() runs forwards inside)
            self._was_persist = [u.persist for u in self.units]
            for u in self.units:
                self.opt._launch_gather(u)
                u.persist = True
            for u in self.units:
                self.opt._make_available(u)
            return self

        def __exit__(self, *exc):
            with torch.no_grad():
                for u, was in zip(self.units, self._was_persist):
                    u.persist = was
                    if u.full is None:
                        continue
                    # write back ONLY under modifier_rank (reference
                    # GatheredParameters semantics) — an unconditional
                    # copy would round-trip weights through the lossy
                    # int8 gather when qwZ is on
                    if self.modifier_rank is not None:
                        if self.opt.world_size > 1:
                            dist.broadcast(u.full, src=self.modifier_rank,
                                           group=self.opt.dp_group)
                        if u.full is not u.shard:  # ws=1 aliases them
                            u.shard.copy_(
                                u.full[self.opt.rank * u.shard_size:
                                       (self.opt.rank + 1) * u.shard_size])
                        if u.sec_shard is not None:
                            ss = u.numel // self.opt.hpz_world
                            u.sec_shard.copy_(
                                u.full[self.opt.hpz_rank * ss:
                                       (self.opt.hpz_rank + 1) * ss])
                    if not u.persist:
                        self.opt._release(u)
            return False

    def gathered_params(self, params, modifier_rank=None):
        return self._Gathered(self, params, modifier_rank)

    @torch.no_grad()
    def get_full_state_dict(self, dtype=None):
        """Gather a consolidated module state_dict on rank 0 (reference
        _zero3_consolidated_16bit_state_dict, engine.py:3693). Returns None
        on other ranks."""
        out = {} if self.rank == 0 else None
        param_names = {p: n for n, p in self.module.named_parameters()}
        for u in self.units:
            full = torch.empty(u.numel, dtype=self._dtype, device=self._device)
            if self.world_size > 1:
                dist.all_gather_into_tensor(full, u.shard, group=self.dp_group)
            else:
                full.copy_(u.shard)
            if self.rank == 0:
                for p, off in zip(u.params, u.offsets):
                    name = param_names.get(p)
                    if name is not None:
                        t = full[off:off + p.ds_numel].view(p.ds_shape).clone()
                        out[name] = t.to(dtype) if dtype is not None else t
            del full
        if out is not None:
            # buffers are not partitioned; keep state_dict() semantics —
            # non-persistent buffers (e.g. HF rotary inv_freq) stay out
            for mname, mod in self.module.named_modules():
                for bname, b in mod._buffers.items():
                    if b is None or \
                            bname in getattr(mod,
                                             "_non_persistent_buffers_set",
                                             ()):
                        continue
                    out[f"{mname}.{bname}" if mname else bname] = \
                        b.detach().clone()
        return out

    @torch.no_grad()
    def get_fp32_state_dict(self, module=None):
        """Consolidated fp32 master state dict on rank 0."""
        module = module or self.module
        out = {} if self.rank == 0 else None
        param_names = {p: n for n, p in module.named_parameters()}
        nvme_masters = {}
        if self.nvme_offload:
            nvme_masters = {gi: torch.nn.Parameter(self._nvme_flat(gi, "p"))
                            for gi in range(len(self.group_shard_numel))}
        for u in self.units:
            if u.group_idx < 0 or not u.trainable:
                continue
            master = nvme_masters.get(u.group_idx,
                                      self.group_masters[u.group_idx])
            shard = master.data[u.master_offset:
                                u.master_offset + u.shard_size].to(self._device)
            full = torch.empty(u.numel, dtype=torch.float32, device=self._device)
            if self.world_size > 1:
                dist.all_gather_into_tensor(full, shard.contiguous(),
                                            group=self.dp_group)
            else:
                full.copy_(shard)
            if self.rank == 0:
                for p, off in zip(u.params, u.offsets):
                    name = param_names.get(p)
                    if name is not None:
                        out[name] = full[off:off + p.ds_numel].view(
                            p.ds_shape).clone()
            del full
        return out

    # ------------------------------------------------------------ checkpoint

    def state_dict(self):
        if self.nvme_offload:
            ng = len(self.group_shard_numel)
            return {
                "stage": 3,
                "world_size": self.world_size,
                "rank": self.rank,
                "loss_scaler": self.loss_scaler.state_dict(),
                "fp32_flat_groups": [self._nvme_flat(gi, "p")
                                     for gi in range(ng)],
                "nvme_moments": [(self._nvme_flat(gi, "m"),
                                  self._nvme_flat(gi, "v"))
                                 for gi in range(ng)],
                "nvme_step": self._nvme_step_count,
                "layout": self.layout_manifest(),
            }
        return {
            "stage": 3,
            "world_size": self.world_size,
            "rank": self.rank,
            "dtype": str(self._dtype),
            "loss_scaler": self.loss_scaler.state_dict(),
            "fp32_flat_groups": [m.data if m.numel() else m
                                 for m in self.group_masters],
            "base_optimizer_state": self.optimizer.state_dict(),
            # layout manifest for the offline universal-checkpoint converter
            # (checkpoint/universal.py) — same schema as stage 1/2
            "layout": self.layout_manifest(),
        }

    def annotate_param_names(self, module):
        """Stamp parameter names used by the universal layout manifest.
        Expert parameters carry their EP rank in the name: with EP > 1 the
        SAME module path holds DIFFERENT experts on each EP rank, and the
        universal per-param files must not collide (universal resume keeps
        the EP size fixed, like the reference)."""
        from ...parallel import groups as pgroups
        for n, p in module.named_parameters():
            if getattr(p, "allreduce", True) is False and \
                    getattr(p, "group_name", None):
                try:
                    ep_rank = pgroups.get_expert_parallel_rank(p.group_name)
                except Exception:
                    ep_rank = 0
                p._ds_name = f"{n}@ep{ep_rank}"
            else:
                p._ds_name = n

    def layout_manifest(self):
        units = []
        for u in self.units:
            units.append({
                "group_idx": u.group_idx,
                "master_offset": u.master_offset,
                "shard_size": u.shard_size,
                "numel": u.numel,
                "pg_world": self.world_size,
                "pg_rank": self.rank,
                "params": [(getattr(p, "_ds_name", None), off, p.ds_numel,
                            tuple(p.ds_shape))
                           for p, off in zip(u.params, u.offsets)],
            })
        return units

    @torch.no_grad()
    def load_universal_state_dict(self, module, usd):
        """Elastic ZeRO-3 load: slice per-param fp32 universal state into
        this world size's unit shards (mirrors stage12's)."""
        if self.nvme_offload:
            raise NotImplementedError(
                "universal-checkpoint load with offload_optimizer.device="
                "nvme: load on cpu/none offload, re-save, then switch")
        self.annotate_param_names(module)
        for gi, group in enumerate(self.optimizer.param_groups):
            master_p = self.group_masters[gi]
            if master_p.numel() == 0:
                continue
            st = self.optimizer.state.setdefault(master_p, {})
            if "exp_avg" not in st:
                st["exp_avg"] = torch.zeros_like(master_p,
                                                 dtype=torch.float32)
                st["exp_avg_sq"] = torch.zeros_like(master_p,
                                                    dtype=torch.float32)
            st["step"] = usd.get("step", 0)
        for u in self.units:
            if u.group_idx < 0 or not u.trainable:
                continue
            master = self.group_masters[u.group_idx]
            st = self.optimizer.state[master]
            dsts = {"param": master.data, "exp_avg": st["exp_avg"],
                    "exp_avg_sq": st["exp_avg_sq"]}
            lo = self.rank * u.shard_size
            hi = lo + u.shard_size
            for p, off in zip(u.params, u.offsets):
                name = getattr(p, "_ds_name", None)
                if name is None or name not in usd["param"]:
                    continue
                a, z = max(off, lo), min(off + p.ds_numel, hi)
                if a >= z:
                    continue
                for kind, dst in dsts.items():
                    src = usd[kind][name].reshape(-1)
                    dst[u.master_offset + (a - lo):
                        u.master_offset + (z - lo)].copy_(src[a - off:z - off])
        self._copy_masters_to_shards()
        self._refresh_persistent()
        self._refresh_secondary()
        for u in self.units:
            if not u.persist and u.status == AVAILABLE:
                self._release(u)

    def load_state_dict(self, sd, load_optimizer_states=True):
        assert sd["world_size"] == self.world_size, \
            "ZeRO-3 checkpoint reshaping requires the universal checkpoint path"
        ck_dtype = sd.get("dtype")
        if ck_dtype is not None and ck_dtype != str(self._dtype):
            raise ValueError(
                f"ZeRO-3 checkpoint was saved by a {ck_dtype} engine but "
                f"this engine runs {self._dtype}: the flat master layouts "
                "differ — convert through the universal checkpoint instead")
        self.loss_scaler.load_state_dict(sd["loss_scaler"])
        if self.nvme_offload:
            for gi, flat in enumerate(sd["fp32_flat_groups"]):
                self._nvme_load_flat(gi, "p", flat)
            if load_optimizer_states and "nvme_moments" in sd:
                for gi, (m, v) in enumerate(sd["nvme_moments"]):
                    self._nvme_load_flat(gi, "m", m)
                    self._nvme_load_flat(gi, "v", v)
                self._nvme_step_count = sd.get("nvme_step", 0)
            self._refresh_persistent()
            self._refresh_secondary()
            for u in self.units:
                if not u.persist and u.status == AVAILABLE:
                    self._release(u)
            return
        for gi, flat in enumerate(sd["fp32_flat_groups"]):
            if self.group_masters[gi].numel():
                self.group_masters[gi].data.copy_(flat)
        if load_optimizer_states:
            self.optimizer.load_state_dict(sd["base_optimizer_state"])
        self._copy_masters_to_shards()
        self._refresh_persistent()
        self._refresh_secondary()
        # any materialized unit must see the new weights
        for u in self.units:
            if not u.persist and u.status == AVAILABLE:
                self._release(u)
