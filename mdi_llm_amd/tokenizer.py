"""Tokenizer loading: HF ``tokenizer.json`` or SentencePiece ``tokenizer.model``.

Capability parity with the reference tokenizer wrapper
(/root/reference/src/sub/tokenizer.py:12-149): auto-detect backend from the
checkpoint directory, resolve bos/eos from ``tokenizer_config.json`` /
``generation_config.json``, and expose ``encode``/``decode``.
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Optional, Union

import torch

__all__ = ["Tokenizer"]


class Tokenizer:
    def __init__(self, checkpoint_dir: Union[str, Path]) -> None:
        checkpoint_dir = Path(checkpoint_dir)
        if not checkpoint_dir.exists():
            raise NotADirectoryError(f"{checkpoint_dir} does not exist")

        self.bos_id: Optional[int] = None
        self.eos_id: Optional[int] = None
        self.use_bos = False
        self.backend: str

        if (vocab_path := checkpoint_dir / "tokenizer.json").is_file():
            from tokenizers import Tokenizer as HFTokenizer

            self.processor = HFTokenizer.from_file(str(vocab_path))
            self.backend = "huggingface"
            self._resolve_special_hf(checkpoint_dir)
        elif (model_path := checkpoint_dir / "tokenizer.model").is_file():
            from sentencepiece import SentencePieceProcessor

            self.processor = SentencePieceProcessor(model_file=str(model_path))
            self.backend = "sentencepiece"
            self.bos_id = self.processor.bos_id()
            self.eos_id = self.processor.eos_id()
            self.use_bos = True
        else:
            raise NotImplementedError(
                f"no tokenizer.json / tokenizer.model in {checkpoint_dir}"
            )

    # -- special-token resolution (reference tokenizer.py:58-117) ---------
    def _resolve_special_hf(self, checkpoint_dir: Path) -> None:
        cfg_path = checkpoint_dir / "tokenizer_config.json"
        cfg = {}
        if cfg_path.is_file():
            with open(cfg_path, encoding="utf-8") as fp:
                cfg = json.load(fp)
        self.use_bos = self._check_use_bos(cfg, checkpoint_dir)
        bos_token = cfg.get("bos_token")
        if isinstance(bos_token, dict):
            bos_token = bos_token.get("content")
        if bos_token is not None:
            self.bos_id = self.token_to_id(bos_token)
        eos_token = cfg.get("eos_token")
        if isinstance(eos_token, dict):
            eos_token = eos_token.get("content")
        if eos_token is not None:
            self.eos_id = self.token_to_id(eos_token)
        gen_path = checkpoint_dir / "generation_config.json"
        if self.eos_id is None and gen_path.is_file():
            with open(gen_path, encoding="utf-8") as fp:
                gen = json.load(fp)
            eos = gen.get("eos_token_id")
            self.eos_id = eos[0] if isinstance(eos, list) else eos

    @staticmethod
    def _check_use_bos(cfg: dict, checkpoint_dir: Path) -> bool:
        if "add_bos_token" in cfg:
            return bool(cfg["add_bos_token"])
        # prefer the tokenizer.json post-processor signal: a template that
        # emits the bos token means the tokenizer itself prepends BOS
        tok_json = checkpoint_dir / "tokenizer.json"
        bos_token = cfg.get("bos_token")
        if isinstance(bos_token, dict):
            bos_token = bos_token.get("content")
        if tok_json.is_file() and bos_token:
            try:
                with open(tok_json, encoding="utf-8") as fp:
                    post = json.load(fp).get("post_processor") or {}
                blob = json.dumps(post)
                return f'"{bos_token}"' in blob
            except (OSError, ValueError):
                pass
        # LLaMA-family tokenizers default to prepending BOS (a bare
        # PreTrainedTokenizerFast does not — reference tokenizer.py:80-85
        # defaults True only for LlamaTokenizer)
        return cfg.get("tokenizer_class") == "LlamaTokenizer"

    # -- API ---------------------------------------------------------------
    @property
    def vocab_size(self) -> int:
        if self.backend == "huggingface":
            return self.processor.get_vocab_size(with_added_tokens=False)
        return self.processor.vocab_size()

    def token_to_id(self, token: str) -> Optional[int]:
        if self.backend == "huggingface":
            return self.processor.token_to_id(token)
        tid = self.processor.piece_to_id(token)
        return tid if tid >= 0 else None

    def encode(
        self,
        string: str,
        device: Optional[torch.device] = None,
        bos: Optional[bool] = None,
        eos: bool = False,
        max_length: int = -1,
    ) -> torch.Tensor:
        if self.backend == "huggingface":
            tokens = self.processor.encode(string).ids
        else:
            tokens = self.processor.encode(string)
        if bos or (bos is None and self.use_bos):
            if self.bos_id is None:
                raise NotImplementedError("tokenizer has no BOS token")
            if not tokens or tokens[0] != self.bos_id:
                tokens = [self.bos_id] + tokens
        if eos and self.eos_id is not None:
            tokens = tokens + [self.eos_id]
        if max_length > 0:
            tokens = tokens[:max_length]
        return torch.tensor(tokens, dtype=torch.int64, device=device)

    def decode(self, tensor: torch.Tensor) -> str:
        tokens = [tensor.item()] if tensor.ndim == 0 else tensor.tolist()
        return self.processor.decode(tokens)
