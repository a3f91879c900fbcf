"""DistriSDXLPipeline / DistriSDPipeline — the user-facing API.

Parity with the reference surface (reference pipelines.py): construction via
``from_pretrained(distri_config, **kw)``, generation via ``pipeline(prompt=
..., generator=...)``, ``prepare()`` runs the comm-buffer registration pass,
the buffer-filling pre-run, and hipGraph capture. Unlike the reference
(which wrapped diffusers' StableDiffusion*Pipeline), the scheduler loop,
prompt encoding, latent handling and VAE decode here are all owned natively.

No network in this environment: ``pretrained_model_name_or_path=None`` (the
default) builds random-init weights of the exact architecture; a local
diffusers-layout directory with safetensors files is loaded when given.
"""

from __future__ import annotations

import json
import os

import torch

from .models.clip import CLIP_VIT_L, OPEN_CLIP_BIG_G, TINY_CLIP, CLIPTextEncoder
from .models.distri_unet import DistriUNet
from .models.tokenizer import CLIPBPETokenizer, SimpleTokenizer, load_clip_tokenizer
from .models.unet import SD15_UNET, SDXL_UNET, TINY_UNET, UNetConfig
from .models.vae import SD_VAE, SDXL_VAE, TINY_VAE, VAEDecoder
from .models import weights as weight_io
from .schedulers import get_scheduler
from .utils.comm import PatchParallelismCommManager
from .utils.config import DistriConfig


def _resolve_tokenizer(pretrained, vocab_size: int, subfolder: str = "tokenizer",
                       pad_with_zero: bool = False, allow_simple: bool = False):
    """Real-checkpoint tokenizer policy: a checkpoint directory
    must ship CLIP BPE files (``<subfolder>/vocab.json`` + ``merges.txt``) —
    hash-token ids against real weights silently break text conditioning.
    Our own ``save_pretrained`` leaves a SimpleTokenizer marker so random-init
    round-trips keep working; anything else must opt in explicitly."""
    if pretrained is None:
        return SimpleTokenizer(vocab_size=vocab_size)
    tok = load_clip_tokenizer(pretrained, subfolder, pad_with_zero=pad_with_zero)
    if tok is not None:
        return tok
    marker = os.path.join(pretrained, subfolder, "tokenizer_config.json")
    marker_simple = False
    if os.path.isfile(marker):
        try:
            with open(marker, encoding="utf-8") as f:
                marker_simple = json.load(f).get("tokenizer_class") == "SimpleTokenizer"
        except (OSError, ValueError):
            pass
    if marker_simple or allow_simple:
        return SimpleTokenizer(vocab_size=vocab_size)
    raise FileNotFoundError(
        f"checkpoint {pretrained!r} has no {subfolder}/vocab.json+merges.txt: pairing real "
        "weights with the hash SimpleTokenizer would feed garbage token ids to the text "
        "encoders (conditioning silently broken). Ship the CLIP BPE files, or pass "
        "allow_simple_tokenizer=True for random-init/debug checkpoints.")


def _maybe_tqdm(iterable, enabled: bool):
    if not enabled:
        return iterable
    try:
        from tqdm import tqdm

        return tqdm(iterable)
    except ImportError:  # pragma: no cover
        return iterable


class _DistriPipelineBase:
    is_sdxl = False

    def __init__(
        self,
        distri_config: DistriConfig,
        unet: DistriUNet,
        vae: VAEDecoder,
        scheduler,
        tokenizer: SimpleTokenizer,
    ):
        self.distri_config = distri_config
        self.unet = unet
        self.vae = vae
        self.scheduler = scheduler
        self.tokenizer = tokenizer
        self.comm_manager: PatchParallelismCommManager | None = None
        self._progress = distri_config.rank == 0 and distri_config.verbose
        self.prepare()

    # -- reference-API conveniences -----------------------------------------

    def set_progress_bar_config(self, disable: bool = False, **kwargs):
        self._progress = not disable and self.distri_config.rank == 0

    def save_pretrained(self, out_dir: str) -> None:
        """Write a diffusers-layout safetensors directory loadable by
        ``from_pretrained(pretrained_model_name_or_path=out_dir)``."""
        import os

        from safetensors.torch import save_file

        from .models.weights import export_diffusers_state_dict

        components = {"unet": self.unet.unet, "vae": self.vae}
        if hasattr(self, "text_encoder"):
            components["text_encoder"] = self.text_encoder
        if hasattr(self, "text_encoder_2"):
            components["text_encoder_2"] = self.text_encoder_2
        for name, model in components.items():
            os.makedirs(os.path.join(out_dir, name), exist_ok=True)
            sd = {k: v.contiguous().cpu() for k, v in export_diffusers_state_dict(model).items()}
            # diffusers repo layout: unet/vae use diffusion_pytorch_model.*,
            # text encoders use model.* (ADVICE r1 — keeps the export loadable
            # by upstream diffusers, not just our own from_pretrained).
            fname = ("diffusion_pytorch_model.safetensors"
                     if name in ("unet", "vae") else "model.safetensors")
            save_file(sd, os.path.join(out_dir, name, fname))
        tokenizers = {"tokenizer": self.tokenizer}
        if getattr(self, "tokenizer_2", None) is not None:
            tokenizers["tokenizer_2"] = self.tokenizer_2
        for sub, tok in tokenizers.items():
            tdir = os.path.join(out_dir, sub)
            os.makedirs(tdir, exist_ok=True)
            with open(os.path.join(tdir, "tokenizer_config.json"), "w", encoding="utf-8") as f:
                json.dump({"tokenizer_class": type(tok).__name__}, f)
            if isinstance(tok, CLIPBPETokenizer):
                import shutil
                shutil.copy(tok.vocab_path, os.path.join(tdir, "vocab.json"))
                shutil.copy(tok.merges_path, os.path.join(tdir, "merges.txt"))

    @property
    def device(self):
        return self.distri_config.device

    # -- core helpers --------------------------------------------------------

    def _unet_dtype(self) -> torch.dtype:
        return next(self.unet.parameters()).dtype

    def _prepare_latents(self, batch_size: int, generator) -> torch.Tensor:
        cfg = self.distri_config
        shape = (batch_size, self.unet.config.in_channels, cfg.height // 8, cfg.width // 8)
        if generator is None:
            # Cross-rank determinism: every rank MUST sample identical initial
            # latents (they denoise the same image). Default to a fixed seed.
            generator = torch.Generator().manual_seed(0)
        device = generator.device if hasattr(generator, "device") else torch.device("cpu")
        latents = torch.randn(shape, generator=generator, dtype=torch.float32, device=device)
        latents = latents.to(device=cfg.device, dtype=self._unet_dtype())
        return latents * self.scheduler.init_noise_sigma

    def _denoise(self, latents, prompt_embeds, added_cond_kwargs, num_inference_steps, guidance_scale):
        cfg = self.distri_config
        do_cfg = cfg.do_classifier_free_guidance
        from .utils.tracing import trace_range

        self.scheduler.set_timesteps(num_inference_steps, device=None)
        self.unet.set_counter(0)
        for i, t in enumerate(_maybe_tqdm(self.scheduler.timesteps, self._progress)):
            with trace_range(f"denoise_step_{i}"):
                latent_in = torch.cat([latents] * 2) if do_cfg else latents
                latent_in = self.scheduler.scale_model_input(latent_in, t)
                noise = self.unet(
                    latent_in,
                    t.to(cfg.device) if torch.is_tensor(t) else t,
                    prompt_embeds,
                    added_cond_kwargs,
                )
                if do_cfg and hasattr(self.scheduler, "guided_step"):
                    latents = self.scheduler.guided_step(noise, t, latents, guidance_scale)
                else:
                    if do_cfg:
                        n_uncond, n_cond = noise.chunk(2)
                        noise = n_uncond + guidance_scale * (n_cond - n_uncond)
                    latents = self.scheduler.step(noise, t, latents)
        return latents

    def _decode(self, latents: torch.Tensor, output_type: str):
        # High-resolution decode: the VAE mid-attention is O(L^2) in latent
        # tokens, so >= 2048^2 outputs use tiled decode automatically
        # (diffusers enable_tiling parity; ~0.7 s at 3840^2 vs ~30 min full)
        if latents.shape[-1] >= 256 or latents.shape[-2] >= 256:
            self.vae.enable_tiling()
        if output_type == "latent":
            return latents
        images = self.vae.decode(latents.to(self._unet_dtype()))
        if output_type == "pt":
            return images
        arr = ((images.float() / 2 + 0.5).clamp(0, 1) * 255).round().to(torch.uint8)
        arr = arr.permute(0, 2, 3, 1).cpu().numpy()
        if output_type == "np":
            return arr
        try:  # "pil"
            from PIL import Image

            return [Image.fromarray(a) for a in arr]
        except ImportError:
            return arr  # no pillow in this environment; numpy HWC uint8

    # -- prepare: registration pass, comm buffer, pre-run, hipGraph capture --

    @torch.no_grad()
    def prepare(self):
        cfg = self.distri_config
        unet = self.unet

        if cfg.device.type == "cuda":
            # MIOpen find mode for whichever convs still ride torch ops: the
            # immediate-mode default picks im2col paths at large resolutions
            # (profiles/rocprof_3840_r01.md). Applies to every entry point,
            # not just bench.py.
            torch.backends.cudnn.benchmark = True

        needs_comm = cfg.parallelism == "patch" and cfg.n_device_per_batch > 1
        wants_graphs = cfg.use_cuda_graph and cfg.device.type == "cuda"
        if not needs_comm and not wants_graphs:
            return  # nothing to warm up
        batch_size = 2 if cfg.do_classifier_free_guidance else 1

        static = self._build_static_inputs(batch_size)

        if needs_comm:
            self.comm_manager = PatchParallelismCommManager(cfg)
            unet.set_comm_manager(self.comm_manager)
            # registration pass: sizes every stale-activation slot
            unet.set_counter(0)
            unet(**static, record=True)
            self.comm_manager.create_buffer()

        # pre-run: modules grab their buffer views and fill them
        unet.set_counter(0)
        unet(**static, record=True)
        if self.comm_manager is not None:
            self.comm_manager.clear()

        if cfg.use_cuda_graph and cfg.device.type == "cuda":
            self._capture_graphs(static)

    def _graph_counters(self) -> list[int]:
        cfg = self.distri_config
        if cfg.world_size == 1 or cfg.n_device_per_batch == 1:
            return [0]
        if cfg.parallelism == "patch":
            return [0, cfg.warmup_steps + 1, cfg.warmup_steps + 2]
        if cfg.parallelism == "naive_patch" and cfg.split_scheme == "alternate":
            return [0, 1]
        return [0]

    def _capture_graphs(self, static):
        import torch.distributed as dist

        unet = self.unet
        cfg = self.distri_config
        graphs, outputs = [], []
        ok = True
        try:
            torch.cuda.synchronize()
            for counter in self._graph_counters():
                g = torch.cuda.CUDAGraph()
                unet.set_counter(counter)
                with torch.cuda.graph(g):
                    out = unet(**static, record=True)
                graphs.append(g)
                outputs.append(out)
        except Exception as exc:  # pragma: no cover - depends on RCCL graph support
            ok = False
            if cfg.rank == 0:
                print(f"[distrifuser_amd] hipGraph capture failed ({exc}); running eager")
        # Consensus: if ANY rank failed to capture, all ranks must run eager,
        # otherwise collective counts diverge and the job deadlocks.
        if cfg.world_size > 1 and dist.is_initialized():
            flag = torch.tensor([0 if ok else 1], device=cfg.device)
            dist.all_reduce(flag, op=dist.ReduceOp.MAX)
            ok = flag.item() == 0
        if ok:
            unet.setup_cuda_graph(outputs, graphs)
        else:
            unet.setup_cuda_graph(None, None)
            if self.comm_manager is not None:
                self.comm_manager.clear()

    def _build_static_inputs(self, batch_size: int) -> dict:
        raise NotImplementedError


class DistriSDXLPipeline(_DistriPipelineBase):
    is_sdxl = True

    def __init__(self, distri_config, unet, vae, text_encoder, text_encoder_2, scheduler,
                 tokenizer, tokenizer_2=None):
        self.text_encoder = text_encoder
        self.text_encoder_2 = text_encoder_2
        self.tokenizer_2 = tokenizer_2
        super().__init__(distri_config, unet, vae, scheduler, tokenizer)

    @staticmethod
    def from_pretrained(distri_config: DistriConfig, **kwargs):
        pretrained = kwargs.pop("pretrained_model_name_or_path", None)
        torch_dtype = kwargs.pop("torch_dtype", torch.bfloat16)
        scheduler = kwargs.pop("scheduler", "ddim")
        preset = kwargs.pop("preset", "sdxl")
        allow_simple = kwargs.pop("allow_simple_tokenizer", False)
        device = distri_config.device

        if preset == "tiny":
            unet_cfg, vae_cfg = TINY_UNET, TINY_VAE
            clip1_cfg, clip2_cfg = TINY_CLIP, TINY_CLIP
        else:
            unet_cfg, vae_cfg = SDXL_UNET, SDXL_VAE
            clip1_cfg, clip2_cfg = CLIP_VIT_L, OPEN_CLIP_BIG_G
        tokenizer = _resolve_tokenizer(pretrained, clip1_cfg.vocab_size,
                                       allow_simple=allow_simple)
        # SDXL's second tokenizer (open-CLIP) pads with 0 ("!"), not eos.
        tokenizer_2 = _resolve_tokenizer(pretrained, clip2_cfg.vocab_size,
                                         subfolder="tokenizer_2", pad_with_zero=True,
                                         allow_simple=allow_simple)
        unet_cfg = kwargs.pop("unet_config", unet_cfg)

        unet = DistriUNet(unet_cfg, distri_config)
        vae = VAEDecoder(vae_cfg)
        te1 = CLIPTextEncoder(clip1_cfg)
        te2 = CLIPTextEncoder(clip2_cfg)

        if pretrained is not None:
            for model, comp in ((unet.unet, "unet"), (vae, "vae"),
                                (te1, "text_encoder"), (te2, "text_encoder_2")):
                path = weight_io.find_component_weights(pretrained, comp)
                if path is not None:
                    weight_io.load_into(model, weight_io.load_safetensors(path))

        unet = unet.to(device=device, dtype=torch_dtype).eval()
        vae = vae.to(device=device, dtype=torch_dtype).eval()
        te1 = te1.to(device=device, dtype=torch_dtype).eval()
        te2 = te2.to(device=device, dtype=torch_dtype).eval()
        return DistriSDXLPipeline(distri_config, unet, vae, te1, te2,
                                  get_scheduler(scheduler), tokenizer,
                                  tokenizer_2=tokenizer_2)

    @torch.no_grad()
    def encode_prompt(self, prompt, negative_prompt=None, do_classifier_free_guidance=True):
        """Returns (prompt_embeds [B,77,2048], pooled [B,1280]) per branch,
        concatenated [uncond; cond] when CFG is on."""
        device = self.distri_config.device
        tok2 = self.tokenizer_2 if self.tokenizer_2 is not None else self.tokenizer
        ids1 = self.tokenizer(prompt, device=device)
        ids2 = tok2(prompt, device=device)
        emb1, _ = self.text_encoder(ids1, hidden_state_index=-2)
        emb2, pooled = self.text_encoder_2(ids2, hidden_state_index=-2)
        embeds = torch.cat([emb1, emb2], dim=-1)
        if not do_classifier_free_guidance:
            return embeds, pooled
        neg = negative_prompt if negative_prompt is not None else ""
        nids = self.tokenizer(neg, device=device)
        nids2 = tok2(neg, device=device)
        nemb1, _ = self.text_encoder(nids, hidden_state_index=-2)
        nemb2, npooled = self.text_encoder_2(nids2, hidden_state_index=-2)
        nembeds = torch.cat([nemb1, nemb2], dim=-1)
        return torch.cat([nembeds, embeds]), torch.cat([npooled, pooled])

    def _added_cond(self, batch_size: int, pooled: torch.Tensor) -> dict:
        cfg = self.distri_config
        time_ids = torch.tensor(
            [[cfg.height, cfg.width, 0, 0, cfg.height, cfg.width]],
            device=cfg.device, dtype=pooled.dtype,
        ).repeat(batch_size, 1)
        return {"text_embeds": pooled, "time_ids": time_ids}

    def _build_static_inputs(self, batch_size: int) -> dict:
        embeds, pooled = self.encode_prompt("", do_classifier_free_guidance=False)
        embeds = embeds.repeat(batch_size, 1, 1).to(self._unet_dtype())
        pooled = pooled.repeat(batch_size, 1).to(self._unet_dtype())
        latents = self._prepare_latents(batch_size, None)
        return {
            "sample": latents,
            "timestep": torch.zeros((), dtype=torch.float32, device=self.distri_config.device),
            "encoder_hidden_states": embeds,
            "added_cond_kwargs": self._added_cond(batch_size, pooled),
        }

    @torch.no_grad()
    def __call__(
        self,
        prompt: str = "",
        negative_prompt: str | None = None,
        num_inference_steps: int = 50,
        guidance_scale: float = 5.0,
        generator: torch.Generator | None = None,
        output_type: str = "pil",
        **kwargs,
    ):
        assert "height" not in kwargs and "width" not in kwargs, (
            "height/width are fixed by DistriConfig (the comm buffers and "
            "graphs are sized for them)"
        )
        cfg = self.distri_config
        if not cfg.do_classifier_free_guidance:
            assert guidance_scale == 1 or guidance_scale is None
            guidance_scale = 1
        do_cfg = cfg.do_classifier_free_guidance

        embeds, pooled = self.encode_prompt(prompt, negative_prompt, do_cfg)
        embeds = embeds.to(self._unet_dtype())
        pooled = pooled.to(self._unet_dtype())
        added = self._added_cond(embeds.shape[0], pooled)
        latents = self._prepare_latents(1, generator)
        latents = self._denoise(latents, embeds, added, num_inference_steps, guidance_scale)
        return self._decode(latents, output_type)


class DistriSDPipeline(_DistriPipelineBase):
    """SD 1.x/2.x pipeline (single text encoder, no added conditions)."""

    def __init__(self, distri_config, unet, vae, text_encoder, scheduler, tokenizer):
        self.text_encoder = text_encoder
        super().__init__(distri_config, unet, vae, scheduler, tokenizer)

    @staticmethod
    def from_pretrained(distri_config: DistriConfig, **kwargs):
        pretrained = kwargs.pop("pretrained_model_name_or_path", None)
        torch_dtype = kwargs.pop("torch_dtype", torch.bfloat16)
        scheduler = kwargs.pop("scheduler", "ddim")
        preset = kwargs.pop("preset", "sd15")
        allow_simple = kwargs.pop("allow_simple_tokenizer", False)
        device = distri_config.device

        if preset == "tiny":
            unet_cfg = UNetConfig(
                block_out_channels=(32, 64),
                down_block_types=("CrossAttnDownBlock2D", "DownBlock2D"),
                layers_per_block=1,
                transformer_layers_per_block=(1, 1),
                num_attention_heads=(2, 4),
                cross_attention_dim=16,
                norm_num_groups=8,
                use_linear_projection=False,
                addition_embed_type=None,
                sample_size=8,
            )
            vae_cfg, clip_cfg = TINY_VAE, TINY_CLIP
        elif preset in ("sd21", "sd2"):
            from .models.clip import OPEN_CLIP_VIT_H
            from .models.unet import SD21_UNET

            unet_cfg, vae_cfg, clip_cfg = SD21_UNET, SD_VAE, OPEN_CLIP_VIT_H
        else:
            unet_cfg, vae_cfg, clip_cfg = SD15_UNET, SD_VAE, CLIP_VIT_L
        tokenizer = _resolve_tokenizer(pretrained, clip_cfg.vocab_size,
                                       allow_simple=allow_simple)
        unet_cfg = kwargs.pop("unet_config", unet_cfg)

        unet = DistriUNet(unet_cfg, distri_config)
        vae = VAEDecoder(vae_cfg)
        te = CLIPTextEncoder(clip_cfg)
        if pretrained is not None:
            for model, comp in ((unet.unet, "unet"), (vae, "vae"), (te, "text_encoder")):
                path = weight_io.find_component_weights(pretrained, comp)
                if path is not None:
                    weight_io.load_into(model, weight_io.load_safetensors(path))

        unet = unet.to(device=device, dtype=torch_dtype).eval()
        vae = vae.to(device=device, dtype=torch_dtype).eval()
        te = te.to(device=device, dtype=torch_dtype).eval()
        return DistriSDPipeline(distri_config, unet, vae, te, get_scheduler(scheduler), tokenizer)

    @torch.no_grad()
    def encode_prompt(self, prompt, negative_prompt=None, do_classifier_free_guidance=True):
        device = self.distri_config.device
        ids = self.tokenizer(prompt, device=device)
        emb, _ = self.text_encoder(ids, hidden_state_index=-1)
        if not do_classifier_free_guidance:
            return emb
        neg = negative_prompt if negative_prompt is not None else ""
        nids = self.tokenizer(neg, device=device)
        nemb, _ = self.text_encoder(nids, hidden_state_index=-1)
        return torch.cat([nemb, emb])

    def _build_static_inputs(self, batch_size: int) -> dict:
        embeds = self.encode_prompt("", do_classifier_free_guidance=False)
        embeds = embeds.repeat(batch_size, 1, 1).to(self._unet_dtype())
        latents = self._prepare_latents(batch_size, None)
        return {
            "sample": latents,
            "timestep": torch.zeros((), dtype=torch.float32, device=self.distri_config.device),
            "encoder_hidden_states": embeds,
            "added_cond_kwargs": None,
        }

    @torch.no_grad()
    def __call__(
        self,
        prompt: str = "",
        negative_prompt: str | None = None,
        num_inference_steps: int = 50,
        guidance_scale: float = 7.5,
        generator: torch.Generator | None = None,
        output_type: str = "pil",
        **kwargs,
    ):
        assert "height" not in kwargs and "width" not in kwargs
        cfg = self.distri_config
        if not cfg.do_classifier_free_guidance:
            guidance_scale = 1
        do_cfg = cfg.do_classifier_free_guidance
        embeds = self.encode_prompt(prompt, negative_prompt, do_cfg).to(self._unet_dtype())
        latents = self._prepare_latents(1, generator)
        latents = self._denoise(latents, embeds, None, num_inference_steps, guidance_scale)
        return self._decode(latents, output_type)
