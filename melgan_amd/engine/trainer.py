"""Training engine — the hot loop.

Per step:
    fake_sub = G(mel)                          # fused HIP convs
    fake     = PQMF.synthesis(fake_sub)        # fused K9
    stft losses (full on fake, sub-band on fake_sub vs PQMF.analysis(audio))
    [after discriminator_train_start_steps] adversarial + feature matching
    g backward  -> bucketed RCCL all-reduce overlapped (parallel.dp)
    g step (fused multi-tensor Adam)
    d loss/backward/step
    periodic: log / validate / checkpoint

bf16 autocast on the model forward; all loss math fp32.
"""

from __future__ import annotations

import contextlib
import itertools
import time

import torch

from melgan_amd.engine import losses as L
from melgan_amd.engine.checkpoint import load_checkpoint, save_checkpoint
from melgan_amd.engine.writer import MetricsWriter, get_logger
from melgan_amd.models import build_discriminator, build_generator
from melgan_amd.ops.pqmf import PQMF
from melgan_amd.ops.stft_loss import MultiResolutionSTFTLoss
from melgan_amd.parallel.dp import BucketAllReduceDP
from melgan_amd.config import check_hp_compat
from melgan_amd.utils.profiling import PhaseTimer


def make_optimizers(hp, generator, discriminator, device):
    from melgan_amd.ops.adam import FusedAdam

    lr = hp.train.adam.lr
    betas = (hp.train.adam.beta1, hp.train.adam.beta2)
    use_fused = hp.backend.fused_adam and device.type == "cuda"
    if use_fused:
        og = FusedAdam(generator.parameters(), lr=lr, betas=betas)
        od = FusedAdam(discriminator.parameters(), lr=lr, betas=betas)
    else:
        og = torch.optim.Adam(generator.parameters(), lr=lr, betas=betas)
        od = torch.optim.Adam(discriminator.parameters(), lr=lr, betas=betas)
    return og, od


class Trainer:
    def __init__(self, hp, name: str, rank: int = 0, world_size: int = 1,
                 device: torch.device | None = None, enable_tb: bool = True):
        self.hp = hp
        self.name = name
        self.rank = rank
        self.world = world_size
        if device is None:
            device = torch.device("cuda", rank % max(
                torch.cuda.device_count(), 1)) \
                if torch.cuda.is_available() else torch.device("cpu")
        self.device = device
        if hp.backend.get("debug_sync"):
            # deterministic launch-order debugging:
            # serialise kernel launches and make HIP errors synchronous
            import os as _os

            _os.environ.setdefault("HIP_LAUNCH_BLOCKING", "1")
            _os.environ.setdefault("AMD_SERIALIZE_KERNEL", "3")
        if hp.backend.get("use_hip_kernels") is False:
            # debugging knob: force the PyTorch composite path even on GPU
            # (same switch the numerics tests use)
            import os as _os

            _os.environ["MELGAN_AMD_DISABLE_HIP"] = "1"

        self.generator = build_generator(hp).to(device)
        self.discriminator = build_discriminator(hp).to(device)
        self.multiband = hp.model.out_channels > 1
        if self.multiband:
            self.pqmf = PQMF(hp.model.out_channels,
                             cutoff_ratio=hp.model.get("pqmf_cutoff")).to(device)
        self.stft_full = MultiResolutionSTFTLoss(
            resolutions=hp.train.stft_loss_params, subband=False).to(device)
        if self.multiband and hp.model.use_subband_stft_loss:
            # family configs may carry their own sub-band resolution set
            # (train.subband_stft_loss_params, dict-of-lists or triples)
            self.stft_sub = MultiResolutionSTFTLoss(
                resolutions=hp.train.get("subband_stft_loss_params"),
                subband=True).to(device)
        else:
            self.stft_sub = None

        self.optim_g, self.optim_d = make_optimizers(
            hp, self.generator, self.discriminator, device)
        self.step = 0

        self.dp_g = BucketAllReduceDP(self.generator,
                                      hp.backend.bucket_cap_mb)
        self.dp_d = BucketAllReduceDP(self.discriminator,
                                      hp.backend.bucket_cap_mb)

        self.amp_dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
                          "none": None, "fp32": None}[hp.backend.amp or "none"]
        self.use_amp = (self.amp_dtype is not None
                        and self.device.type == "cuda")
        self.logger = get_logger(hp.log.log_dir, name) if rank == 0 else None
        if self.amp_dtype is torch.float16 and self.logger:
            self.logger.warning(
                "backend.amp=fp16 compiles but is UNTESTED on GPU (no loss "
                "scaling; bf16 is the production dtype — NOTES.md)")
        if self.logger:
            from melgan_amd.config import unknown_keys

            unk = unknown_keys(hp)
            if unk:
                self.logger.warning(
                    f"config keys the framework never reads (typo?): {unk}")
        self.writer = MetricsWriter(hp.log.log_dir, name,
                                    enable_tb=enable_tb, rank=rank)
        self.lambda_adv = hp.train.get("lambda_adv") or 2.5
        self.lambda_feat = hp.model.feat_match
        self.timer = PhaseTimer(enabled=bool(hp.backend.get("phase_timing"))
                                and self.device.type == "cuda")
        self.last_phases = {}
        # side HIP stream for the multi-resolution STFT losses.  OFF BY
        # DEFAULT: a 2000-step soak NaNs at step ~600-1600 with it on
        # (reproduced with loss_stream alone; d_stream alone and both-off
        # run 2000 steps clean), and record_stream + a pre-optimizer
        # wait_stream did not cure it — root cause still open (NOTES.md).
        # Opt in with backend: {loss_stream: true}.
        use_ls = hp.backend.get("loss_stream")
        self.loss_stream = torch.cuda.Stream() \
            if use_ls and self.device.type == "cuda" else None
        # the d-phase forward runs on its own stream so its BACKWARD
        # (replayed there by autograd) overlaps the generator backward
        use_ds = hp.backend.get("d_stream")
        self.d_stream = torch.cuda.Stream() \
            if (use_ds is None or use_ds) and self.device.type == "cuda" \
            else None

    # ------------------------------------------------------------------
    def load(self, chkpt_path: str):
        ck = load_checkpoint(chkpt_path, map_location=self.device)
        from melgan_amd.engine.checkpoint import load_flexible

        load_flexible(self.generator, ck["model_g"], logger=self.logger)
        load_flexible(self.discriminator, ck["model_d"], logger=self.logger)
        try:
            self.optim_g.load_state_dict(ck["optim_g"])
            self.optim_d.load_state_dict(ck["optim_d"])
        except Exception:
            if self.logger:
                self.logger.warning("optimizer state incompatible; "
                                    "reinitialised")
        self.step = ck.get("step", 0)
        if "hp_str" in ck:
            check_hp_compat(ck["hp_str"], self.hp.hp_str, self.logger)
        if self.logger:
            self.logger.info(f"resumed from {chkpt_path} at step {self.step}")

    def save(self):
        if self.rank != 0:
            return None
        return save_checkpoint(
            self.hp.log.chkpt_dir, self.name, self.step, self.generator,
            self.discriminator, self.optim_g, self.optim_d, self.hp.hp_str)

    # ------------------------------------------------------------------
    def _prewarm_disc(self):
        """Materialise the discriminator's per-step weight folds and GEMM
        operand caches on the CURRENT stream, so the d-stream fork and
        the main-stream frozen forward can consume them concurrently
        without a cross-stream race on first use."""
        from melgan_amd.ops import _ext, fused

        if _ext.ext() is None or self.device.type != "cuda":
            return
        dt = self.amp_dtype or torch.float32
        for m in self.discriminator.modules():
            if hasattr(m, "weight_v") and not getattr(m, "_removed", True):
                w_c, _ = fused._wn_cached(m, dt)
                if fused._gemm_fwd_eligible(
                        w_c, m.in_channels, m.out_channels, m.kernel_size,
                        m.stride, m.groups):
                    fused._apad_cached(m, w_c)
                if (m.stride == 1 and m.groups == 1 and
                        fused._gemm_bwd_eligible(dt, m.in_channels,
                                                 m.out_channels,
                                                 m.kernel_size)):
                    fused._apad_bwd_cached(m, w_c)

    def _stft_losses(self, fake_sub, fake, audio):
        """Multi-resolution STFT losses (full-band + sub-band mix)."""
        sc_full, mag_full = self.stft_full(fake.squeeze(1),
                                           audio.squeeze(1).float())
        stft_loss = sc_full + mag_full
        sub_loss = None
        if self.stft_sub is not None:
            with torch.no_grad():
                audio_sub = self.pqmf.analysis(audio.float())
            sc_sub, mag_sub = self.stft_sub(fake_sub, audio_sub)
            sub_loss = sc_sub + mag_sub
            stft_loss = 0.5 * stft_loss + 0.5 * sub_loss
        return stft_loss, sub_loss

    def _forward_g(self, mel, audio):
        """Generator forward + generator losses (no discriminator reuse —
        kept for the warmup phase and standalone callers). Returns
        (loss, parts, fake full-band audio)."""
        with torch.autocast("cuda", dtype=self.amp_dtype,
                            enabled=self.use_amp):
            fake_sub = self.generator(mel)  # (B, N_bands, S/N)
        fake_sub = fake_sub.float()
        fake = self.pqmf.synthesis(fake_sub) if self.multiband else fake_sub
        stft_loss, sub_loss = self._stft_losses(fake_sub, fake, audio)
        return stft_loss, dict(stft=stft_loss, sub=sub_loss, adv=None,
                               feat=None), fake

    def train_step(self, mel, audio):
        hp = self.hp
        mel = mel.to(self.device, non_blocking=True)
        audio = audio.to(self.device, non_blocking=True)
        adversarial = self.step >= hp.train.discriminator_train_start_steps

        tm = self.timer
        # ---- generator forward + losses --------------------------------
        tm.mark("g_fwd")
        self.dp_g.zero_grad(self.optim_g)
        if adversarial and self.loss_stream is not None:
            # generator forward on the main stream; STFT losses on the
            # side stream while the discriminator phase runs below
            with torch.autocast("cuda", dtype=self.amp_dtype,
                                enabled=self.use_amp):
                fake_sub = self.generator(mel)
            fake_sub = fake_sub.float()
            fake = self.pqmf.synthesis(fake_sub) if self.multiband \
                else fake_sub
            self.loss_stream.wait_stream(torch.cuda.current_stream())
            # main-stream tensors consumed on the side stream: guard the
            # allocator against reuse while side-stream work is in flight
            fake_sub.record_stream(self.loss_stream)
            fake.record_stream(self.loss_stream)
            audio.record_stream(self.loss_stream)
            with torch.cuda.stream(self.loss_stream):
                stft_loss, sub_loss = self._stft_losses(fake_sub, fake,
                                                        audio)
            g_loss = None  # joined after the discriminator forwards
            parts = dict(stft=stft_loss, sub=sub_loss, adv=None, feat=None)
        else:
            g_loss, parts, fake = self._forward_g(mel, audio)

        # ---- discriminator forwards (both phases share D(real)) --------
        d_loss = None
        if adversarial:
            tm.mark("d_fwd")
            self.dp_d.zero_grad(self.optim_d)
            B = audio.size(0)
            audio_f = audio.float()
            fake_det = fake.detach()
            ds = self.d_stream
            main = torch.cuda.current_stream() if ds is not None else None
            if ds is not None:
                # weight folds / GEMM operand caches must exist BEFORE
                # the fork — both streams' forwards consume them
                self._prewarm_disc()
                ds.wait_stream(main)
            with torch.cuda.stream(ds) if ds is not None \
                    else contextlib.nullcontext():
                with torch.autocast("cuda", dtype=self.amp_dtype,
                                    enabled=self.use_amp):
                    # one 2B-batch forward instead of two B-batch
                    # forwards: identical math, 2x per-kernel tile fill;
                    # it runs on the d stream CONCURRENTLY with the
                    # frozen g-phase forward below, and its backward
                    # replays there, overlapping the generator backward
                    d_both = self.discriminator(
                        torch.cat([audio_f, fake_det], dim=0))

            # Freeze discriminator PARAMS for the generator-phase
            # adversarial forward: the g loss needs gradients THROUGH the
            # discriminator to `fake`, not INTO its weights.  Without
            # this, g_loss.backward() (a) wastes the disc weight-grad
            # kernels and (b) fires the disc's DP bucket hooks with
            # g-phase gradients, which would make the d-step's
            # all-reduce ship STALE gradients at world_size>1.
            tm.mark("g_adv_fwd")
            for p in self.discriminator.parameters():
                p.requires_grad_(False)
            try:
                with torch.autocast("cuda", dtype=self.amp_dtype,
                                    enabled=self.use_amp):
                    d_fake = self.discriminator(fake)
            finally:
                for p in self.discriminator.parameters():
                    p.requires_grad_(True)

            if ds is not None:
                audio_f.record_stream(ds)
                fake_det.record_stream(ds)
                main.wait_stream(ds)
                for feats in d_both:
                    for f in feats:
                        f.record_stream(main)
            d_real = [[f[:B] for f in feats] for feats in d_both]
            d_fake_det = [[f[B:] for f in feats] for feats in d_both]
            d_loss = L.discriminator_loss(d_real, d_fake_det)

            adv = L.adversarial_g_loss(d_fake)
            # feature-matching targets reuse the d-phase D(real) features
            # (detached inside the loss) — D runs on real audio ONCE/step
            feat = L.feature_matching_loss(d_fake, d_real)
            if g_loss is None:  # join the side-stream STFT losses
                torch.cuda.current_stream().wait_stream(self.loss_stream)
                stft = parts["stft"]
                stft.record_stream(torch.cuda.current_stream())
                g_loss = stft
            g_loss = g_loss + self.lambda_adv * adv + self.lambda_feat * feat
            parts["adv"], parts["feat"] = adv, feat

        # ---- generator update (D still at pre-step weights) -------------
        tm.mark("g_bwd")
        g_loss.backward()
        if adversarial and self.loss_stream is not None:
            # the STFT-path backward nodes replayed on the loss stream;
            # their generator-grad contributions must be complete before
            # the fused Adam (raw-pointer tables) reads the grads
            torch.cuda.current_stream().wait_stream(self.loss_stream)
        tm.mark("g_comm_wait")
        self.dp_g.finish()
        tm.mark("g_opt")
        self.optim_g.step()

        # ---- discriminator update ---------------------------------------
        if adversarial:
            tm.mark("d_bwd")
            d_loss.backward()  # tower bwd kernels replay on d_stream,
            # concurrent with the generator backward/step above
            tm.mark("d_comm_wait")
            self.dp_d.finish()
            tm.mark("d_opt")
            if self.d_stream is not None:
                # d grads were produced on d_stream; the fused Adam
                # launches on the main stream
                torch.cuda.current_stream().wait_stream(self.d_stream)
            self.optim_d.step()
        tm.mark(None)
        if tm.enabled:
            self.last_phases = tm.collect()

        self.step += 1
        return g_loss.detach(), parts, \
            d_loss.detach() if d_loss is not None else None

    # ------------------------------------------------------------------
    def train(self, trainloader, valloader=None, max_steps: int | None = None):
        hp = self.hp
        max_steps = max_steps or hp.train.max_steps
        frames_per_batch = (hp.train.batch_size *
                            hp.audio.segment_length // hp.audio.hop_length)
        t0 = time.time()
        steps_t0 = self.step
        sampler = getattr(trainloader, "sampler", None)
        # explicit epoch loop (NOT itertools.cycle, which would memoise the
        # first epoch's random crops forever and pin their memory, and never
        # reshuffle DistributedSampler shards)
        for epoch in itertools.count():
            if self.step >= max_steps:
                break
            if isinstance(sampler,
                          torch.utils.data.distributed.DistributedSampler):
                sampler.set_epoch(epoch)
            saw_batch = False
            for mel, audio in trainloader:
                saw_batch = True
                if self.step >= max_steps:
                    break
                g_loss, parts, d_loss = self.train_step(mel, audio)

                if self.step % hp.log.summary_interval == 0 and \
                        self.rank == 0:
                    if self.device.type == "cuda":
                        torch.cuda.synchronize()
                    dt = time.time() - t0
                    nsteps = self.step - steps_t0
                    mfps = (nsteps * frames_per_batch * self.world / dt) \
                        if dt > 0 else 0.0
                    self.writer.log_training(
                        self.step, g_loss=g_loss, d_loss=d_loss,
                        stft=parts["stft"], sub=parts["sub"],
                        adv=parts["adv"], feat=parts["feat"],
                        mel_frames_per_sec=mfps,
                        **{f"ms_{k}": v
                           for k, v in self.last_phases.items()})
                    if self.logger:
                        self.logger.info(
                            f"step {self.step} g_loss {float(g_loss):.4f}"
                            + (f" d_loss {float(d_loss):.4f}"
                               if d_loss is not None else "")
                            + f" | {mfps:,.0f} mel-frames/s")
                    t0 = time.time()
                    steps_t0 = self.step

                if valloader is not None and \
                        self.step % hp.log.validation_interval == 0:
                    self.validate(valloader)

                if self.step % hp.log.save_interval == 0:
                    self.save()
            if not saw_batch:  # exhausted one-shot iterable: stop
                break
        return self.step

    @torch.no_grad()
    def validate(self, valloader, max_batches: int = 8):
        self.generator.eval()
        tot_sc = tot_mag = 0.0
        n = 0
        last_audio = None
        for i, (mel, audio) in enumerate(valloader):
            if i >= max_batches:
                break
            mel = mel.to(self.device)
            audio = audio.to(self.device)
            fake_sub = self.generator(mel).float()
            fake = self.pqmf.synthesis(fake_sub) if self.multiband else fake_sub
            # full-length val clips: the synthesised length (mel_frames*hop)
            # overshoots the true audio length (mel framing rounds up) —
            # compare on the common prefix
            n_s = min(fake.size(2), audio.size(2))
            sc, mag = self.stft_full(fake[:, :, :n_s].squeeze(1),
                                     audio[:, :, :n_s].squeeze(1).float())
            tot_sc += float(sc)
            tot_mag += float(mag)
            n += 1
            last_audio = fake[0, 0].float().cpu()
        self.generator.train()
        if n and self.rank == 0:
            self.writer.log_validation(
                self.step, audio=last_audio, sr=self.hp.audio.sampling_rate,
                val_sc=tot_sc / n, val_mag=tot_mag / n)
        return (tot_sc / n, tot_mag / n) if n else (0.0, 0.0)
