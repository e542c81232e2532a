"""CaptionGenerator — the whole model graph (parity with reference
`model.py` CaptionGenerator).

Train forward (model.py:190-334): CNN encoder -> context grid -> T-step
attention-LSTM teacher-forced loop -> three-part loss:
  * masked softmax cross-entropy, Σ/Σmasks (model.py:294-297, 316-318);
  * doubly-stochastic attention regularizer
    attention_loss_factor · l2_loss(1 − Σ_t masked α_t) / (B·L)
    (model.py:266-269, 320-326);
  * L2 regularization of fc/conv kernels (model.py:328, nn.py:17-43);
plus masked argmax accuracy (model.py:300-305, 332-334).

Inference surface (model.py:336-354): `encode(images)` = conv_feats +
initial LSTM state (the infer graph's first sess.run, base_model.py:167-170);
`decode_step(...)` = one decoder step returning (memory, output, probs) (the
per-beam-step sess.run, base_model.py:207-212).
"""

import torch
import torch.nn as tnn

from .. import ops
from .decoder import AttentionDecoder
from .encoders import build_encoder
from .nn import NN


class CaptionGenerator(tnn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.is_train = getattr(config, 'phase', 'train') == 'train'
        self.train_cnn = self.is_train and getattr(config, 'train_cnn',
                                                   False)
        self.nn = NN(config)
        self.cnn = build_encoder(config.cnn, self.nn)
        self.num_ctx = self.cnn.num_ctx
        self.dim_ctx = self.cnn.dim_ctx
        self.decoder = AttentionDecoder(config, self.nn,
                                        self.dim_ctx, self.num_ctx)
        self.global_step = 0
        self._cnn_channels_last = False

        # freeze policy: CNN params trainable only with --train_cnn
        # (reference nn.py:66); everything non-trainable at eval/test.
        for p in self.cnn.parameters():
            p.requires_grad_(self.train_cnn)
        for p in self.decoder.parameters():
            p.requires_grad_(self.is_train)

    # ---- encoder ----

    def _compute_dtype(self, images):
        return (torch.bfloat16
                if (images.is_cuda
                    and getattr(self.config, 'compute_dtype', 'bf16')
                    == 'bf16')
                else torch.float32)

    def _use_bptt(self, contexts):
        """The fused BPTT path covers the reference default architecture
        (2-layer attend/decode MLPs) on GPU bf16."""
        cfg = self.config
        return (contexts.is_cuda
                and contexts.dtype == torch.bfloat16
                and getattr(cfg, 'use_hip_kernels', True)
                and getattr(cfg, 'use_bptt', True)
                # activity regularizers need per-layer activations in the
                # autograd graph — the fused BPTT doesn't expose them
                and getattr(cfg, 'fc_activity_regularizer_scale', 0.0)
                == 0.0
                and cfg.num_attend_layers == 2
                and cfg.num_decode_layers == 2
                # kernel shape contracts (fall back to the per-op loop
                # for exotic dims): bf16x8 K-tiles + scores-bwd A chunks
                and cfg.dim_attend_layer % 512 == 0
                and cfg.dim_attend_layer <= 2048
                and self.dim_ctx % 8 == 0
                and cfg.dim_embedding % 8 == 0
                and cfg.num_lstm_units % 8 == 0
                and cfg.dim_decode_layer % 8 == 0
                and self.num_ctx <= 1024)

    def compute_contexts(self, images):
        """images: [B,3,224,224] float -> contexts [B,L,D].

        On GPU the conv stack runs NHWC (channels_last): MIOpen's bf16
        igemm kernels are NHWC-native; NCHW falls back to a naive conv
        (measured 70% of step time, profiles/r01_baseline.md)."""
        dtype = self._compute_dtype(images)
        images = images.to(dtype)
        if images.is_cuda:
            if not self._cnn_channels_last:
                self.cnn.to(memory_format=torch.channels_last)
                self._cnn_channels_last = True
            images = images.contiguous(memory_format=torch.channels_last)
        if self.train_cnn:
            return self.cnn(images)
        with torch.no_grad():
            return self.cnn(images)

    # ---- training ----

    def forward(self, images, sentences, masks):
        """Teacher-forced training forward.

        images: [B,3,224,224], sentences: [B,T] int64, masks: [B,T] float.
        Returns dict of scalar losses + accuracy + attentions.
        """
        cfg = self.config
        B = images.shape[0]
        T = cfg.max_caption_length

        contexts = self.compute_contexts(images)
        contexts_flat = contexts.reshape(-1, self.dim_ctx)
        use_bptt = self._use_bptt(contexts)
        self.decoder.precast(contexts.dtype, skip_bptt=use_bptt)

        context_mean = contexts.float().mean(dim=1).to(contexts.dtype)
        initial_memory, initial_output = self.decoder.initialize(
            context_mean)

        mask_sum = masks.sum()
        if use_bptt:
            # fused hand-written BPTT over all T steps (sat_amd.models.bptt)
            from .bptt import run_decoder_bptt
            ce, attentions, predictions = run_decoder_bptt(
                self.decoder, contexts, initial_memory, initial_output,
                sentences, masks)
            cross_entropy_loss = ce.sum() / mask_sum
            with torch.no_grad():
                num_correct = [((predictions == sentences).float()
                                * masks).sum()]
        else:
            memory, output = initial_memory, initial_output
            state_h = output
            last_word = torch.zeros(B, dtype=torch.int64,
                                    device=images.device)

            cross_entropies = []
            masked_alphas = []
            num_correct = []
            for t in range(T):
                logits, alpha, memory, output, state_h = \
                    self.decoder.step(
                        contexts, contexts_flat, last_word, memory,
                        output, state_h, salt=t)
                m = masks[:, t]
                cross_entropies.append(
                    ops.masked_softmax_ce(logits, sentences[:, t], m))
                masked_alphas.append(alpha.float() * m.unsqueeze(1))
                with torch.no_grad():
                    pred = logits.argmax(dim=1)
                    num_correct.append(((pred == sentences[:, t]).float()
                                        * m).sum())
                last_word = sentences[:, t]

            cross_entropy_loss = torch.stack(
                cross_entropies, dim=1).sum() / mask_sum
            attentions = torch.stack(masked_alphas, dim=2).sum(dim=2)

        diffs = 1.0 - attentions
        attention_loss = cfg.attention_loss_factor \
            * 0.5 * (diffs ** 2).sum() / (B * self.num_ctx)

        reg_loss = self.nn.reg_loss()
        if isinstance(reg_loss, torch.Tensor):
            reg_loss = reg_loss.to(cross_entropy_loss.device)

        self.decoder.clear_cast()
        total_loss = cross_entropy_loss + attention_loss + reg_loss
        accuracy = torch.stack(num_correct).sum() / mask_sum

        return {
            'total_loss': total_loss,
            'cross_entropy_loss': cross_entropy_loss,
            'attention_loss': attention_loss,
            'reg_loss': reg_loss,
            'accuracy': accuracy,
            'attentions': attentions,
        }

    # ---- inference (beam search) ----

    @torch.no_grad()
    def encode(self, images):
        """-> (contexts [B,L,D], initial_memory, initial_output)."""
        contexts = self.compute_contexts(images)
        context_mean = contexts.float().mean(dim=1).to(contexts.dtype)
        memory, output = self.decoder.initialize(context_mean)
        return contexts, memory, output

    @torch.no_grad()
    def decode_step(self, contexts, last_word, last_memory, last_output):
        """One inference decoder step -> (memory, output, probs [B,V])."""
        contexts_flat = contexts.reshape(-1, self.dim_ctx)
        self.decoder.precast(contexts.dtype)
        logits, _alpha, memory, output, _sh = self.decoder.step(
            contexts, contexts_flat, last_word, last_memory, last_output)
        self.decoder.clear_cast()
        probs = torch.softmax(logits.float(), dim=1)
        return memory, output, probs
