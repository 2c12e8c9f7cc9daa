"""Serving: per-client drift-aware prediction from checkpoint files.

The reference has no explicit serving path (evaluation is prequential,
server-side only — SURVEY.md section 3.5); in production the trained
artifact is the pair (model_params.pt, <algo>_state.pkl), and the routing
rule is "client c is served by its cluster's model"
(sc_state.get_test_model_idx / mm_state / ds_state.get_model_key). This
module packages that rule behind one API so a deployment can load a
checkpoint and answer per-client queries on the GPU.
"""

from __future__ import annotations

import os
import pickle
from typing import Optional

import numpy as np
import torch

from ..config import Config
from ..models import packed, zoo
from ..models.generic_packer import ModulePacker


class DriftModelServer:
    def __init__(self, cfg: Config, ckpt_dir: str,
                 device: Optional[torch.device] = None):
        self.cfg = cfg
        self.device = device or torch.device(
            "cuda:0" if torch.cuda.is_available() else "cpu")
        mp = torch.load(os.path.join(ckpt_dir, "model_params.pt"))
        self.n_models = len(mp)
        from ..data.generators import CLASS_NUM, FEATURE_NUM
        ds = "MNIST" if cfg.dataset.lower() == "mnist" else cfg.dataset
        self.class_num = CLASS_NUM[ds]
        self.feature_num = FEATURE_NUM[ds]

        self.is_module = cfg.model not in ("lr", "fnn")
        proto = zoo.create_model(cfg.model, self.class_num,
                                 self.feature_num)
        if self.is_module:
            self.packer = ModulePacker(proto)
            self.module = proto.to(self.device)
        else:
            self.spec = packed.spec_for(cfg.model, self.feature_num,
                                        self.class_num)
            self.packer = packed.PackedMLP(self.spec)
        self.params = torch.stack(
            [self.packer.flatten(mp[m]) for m in sorted(mp)]).to(self.device)

        # per-client routing from the algorithm state
        self.route = self._load_routing(ckpt_dir)

    def _load_routing(self, ckpt_dir: str) -> np.ndarray:
        C = self.cfg.client_num_in_total
        algo = self.cfg.concept_drift_algo

        def pkl(name):
            with open(os.path.join(ckpt_dir, name), "rb") as f:
                return pickle.load(f)

        if self.cfg.is_softcluster:
            st = pkl("sc_state.pkl")
            it = max(st.train_data_weights.keys())
            return np.array([st.get_test_model_idx(it, c)
                             for c in range(C)])
        if algo in ("mmacc", "mmgeni", "mmgeniex"):
            st = pkl("mm_state.pkl")
            return np.array([min(st.get_test_model_idx(c),
                                 self.n_models - 1) for c in range(C)])
        if algo in ("driftsurf", "dsurf"):
            st = pkl("ds_state.pkl")
            idx = 0
            for i, key in enumerate(st.get_train_keys()):
                if key == st.get_model_key():
                    idx = i
            return np.full(C, min(idx, self.n_models - 1))
        return np.zeros(C, dtype=int)   # single-model / ensemble head

    @torch.no_grad()
    def predict(self, client: int, x: np.ndarray) -> np.ndarray:
        """Class predictions for client `client` on rows x [n, D]."""
        m = int(self.route[client])
        xt = torch.as_tensor(np.ascontiguousarray(x, dtype=np.float32),
                             device=self.device)
        if self.is_module:
            self.packer.load_into(self.module, self.params[m])
            self.module.eval()
            logits = self.module(xt)
        else:
            from ..ops import mlp_torch
            logits = mlp_torch.forward_logits(
                self.spec, self.params[m:m + 1], xt.unsqueeze(0)).squeeze(0)
        return logits.argmax(-1).cpu().numpy()
