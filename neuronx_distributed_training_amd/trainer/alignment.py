"""DPO / ORPO model modules (reference base_dpo.py / base_orpo.py parity).

DPO: reference-model log-probs are precomputed in a no-grad pass over the
whole train set at ``on_train_start`` and cached per sample (reference
base_dpo.py:24-66); the policy runs a concatenated forward (chosen +
rejected stacked on the batch dim, :68-88) and a sigmoid DPO loss with
``kl_beta`` + reward metrics (:90-109). TP-aware log-probs via
``from_parallel_logits_to_logprobs``.

ORPO: odds-ratio preference loss, no reference model, average log-probs
(reference base_orpo.py).
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader

from ..parallel.loss import from_parallel_logits_to_logprobs
from .module import LlamaModule


class _WithRefLogps(torch.utils.data.Dataset):
    def __init__(self, base, ref_chosen, ref_rejected):
        self.base = base
        self.ref_chosen = ref_chosen
        self.ref_rejected = ref_rejected

    def __len__(self):
        return len(self.base)

    def __getitem__(self, i):
        s = dict(self.base[i])
        s["ref_chosen_logps"] = self.ref_chosen[i]
        s["ref_rejected_logps"] = self.ref_rejected[i]
        return s


class DPOBaseModel(LlamaModule):
    needs_ref_model = True

    def __init__(self, cfg: Dict):
        super().__init__(cfg)
        dstr = cfg.get("distributed_strategy", {})
        if int(dstr.get("pipeline_model_parallel_size", 1) or 1) > 1 or \
                int(dstr.get("context_parallel_size", 1) or 1) > 1:
            raise ValueError(
                "DPO/ORPO run with TP/DP only (sequence logprobs need the "
                "full model on each PP stage and the full sequence per rank)"
            )
        align = cfg.get("model_alignment_strategy", {})
        dcfg = align.get("dpo") or align.get("orpo") or {}
        self.kl_beta = float(dcfg.get("kl_beta", dcfg.get("beta", 0.1)))

    # -- log-prob of the labeled (response) tokens of each sample --
    def _sequence_logps(self, input_ids, labels, loss_mask, average=False,
                        attention_mask=None):
        hidden = self.model.model(input_ids, attention_mask=attention_mask)
        logits = self.model.lm_head(
            hidden, pre_mapped=self.model.cfg.sequence_parallel
        )  # [b, s, v/tp]
        safe_labels = labels.clamp(min=0)
        lp = from_parallel_logits_to_logprobs(logits, safe_labels)  # [b, s-1]
        m = loss_mask[:, 1:].to(lp.dtype)
        s = (lp * m).sum(-1)
        if average:
            return s / m.sum(-1).clamp(min=1)
        return s

    def on_train_start(self, datamodule):
        if not self.needs_ref_model:
            return
        # precompute frozen-reference logps over the whole train set
        self.model.eval()
        ds = datamodule.train_ds
        loader = DataLoader(ds, batch_size=self.micro_batch_size, shuffle=False)
        ref_c, ref_r = [], []
        with torch.no_grad():
            for batch in loader:
                batch = {
                    k: (v.to(self.device) if torch.is_tensor(v) else v)
                    for k, v in batch.items()
                }
                c = self._sequence_logps(
                    batch["chosen_input_ids"], batch["chosen_labels"],
                    batch["chosen_loss_mask"],
                    attention_mask=batch.get("chosen_attention_mask"),
                )
                r = self._sequence_logps(
                    batch["rejected_input_ids"], batch["rejected_labels"],
                    batch["rejected_loss_mask"],
                    attention_mask=batch.get("rejected_attention_mask"),
                )
                ref_c.append(c.float().cpu())
                ref_r.append(r.float().cpu())
        self.model.train()
        datamodule.train_ds = _WithRefLogps(
            ds, torch.cat(ref_c), torch.cat(ref_r)
        )

    def preference_loss(self, pol_c, pol_r, ref_c, ref_r):
        logits = (pol_c - ref_c) - (pol_r - ref_r)
        loss = -F.logsigmoid(self.kl_beta * logits).mean()
        chosen_reward = self.kl_beta * (pol_c - ref_c).detach().mean()
        rejected_reward = self.kl_beta * (pol_r - ref_r).detach().mean()
        acc = (logits.detach() > 0).float().mean()
        return loss, {
            "rewards_chosen": float(chosen_reward),
            "rewards_rejected": float(rejected_reward),
            "reward_accuracy": float(acc),
        }

    def model_fwd_calc_loss(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        # concatenated forward: chosen + rejected on the batch dim
        ids = torch.cat([batch["chosen_input_ids"], batch["rejected_input_ids"]])
        labels = torch.cat([batch["chosen_labels"], batch["rejected_labels"]])
        mask = torch.cat([batch["chosen_loss_mask"], batch["rejected_loss_mask"]])
        am = None
        if batch.get("chosen_attention_mask") is not None:
            am = torch.cat(
                [batch["chosen_attention_mask"], batch["rejected_attention_mask"]]
            )
        logps = self._sequence_logps(ids, labels, mask, attention_mask=am)
        b = batch["chosen_input_ids"].size(0)
        pol_c, pol_r = logps[:b], logps[b:]
        ref_c = batch["ref_chosen_logps"].to(pol_c.device).to(pol_c.dtype)
        ref_r = batch["ref_rejected_logps"].to(pol_r.device).to(pol_r.dtype)
        loss, metrics = self.preference_loss(pol_c, pol_r, ref_c, ref_r)
        self.last_alignment_metrics = metrics
        return loss

    def training_step(self, microbatches):
        metrics = super().training_step(microbatches)
        metrics.update(getattr(self, "last_alignment_metrics", {}))
        return metrics


class ORPOBaseModel(DPOBaseModel):
    """Odds-ratio preference optimization: no reference model
    (reference base_orpo.py:14-45)."""

    needs_ref_model = False

    def model_fwd_calc_loss(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        ids = torch.cat([batch["chosen_input_ids"], batch["rejected_input_ids"]])
        labels = torch.cat([batch["chosen_labels"], batch["rejected_labels"]])
        mask = torch.cat([batch["chosen_loss_mask"], batch["rejected_loss_mask"]])
        am = None
        if batch.get("chosen_attention_mask") is not None:
            am = torch.cat(
                [batch["chosen_attention_mask"], batch["rejected_attention_mask"]]
            )
        avg_logps = self._sequence_logps(ids, labels, mask, average=True,
                                         attention_mask=am)
        b = batch["chosen_input_ids"].size(0)
        lc, lr = avg_logps[:b], avg_logps[b:]
        # log odds: log(p/(1-p)) with p = exp(avg_logp)
        def log_odds(lp):
            return lp - torch.log1p(-torch.exp(lp).clamp(max=1 - 1e-6))
        ratio = log_odds(lc) - log_odds(lr)
        or_loss = -F.logsigmoid(self.kl_beta * ratio).mean()
        nll = -lc.mean()
        self.last_alignment_metrics = {
            "log_odds_ratio": float(ratio.detach().mean()),
            "sft_nll": float(nll.detach()),
        }
        return nll + or_loss


# entry-point aliases (examples/training.py build_module)
DPOModule = DPOBaseModel
ORPOModule = ORPOBaseModel
