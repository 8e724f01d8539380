"""Interactive chat REPL.

Rebuild of the reference ChatInterface (/root/reference/Src/Main_Scripts/
Chat.py:472-937): checkpoint auto-discovery, architecture inference,
conversation history windowing, commands /help /stats /mode /system /save
/config /clear /quit — on top of the KV-cached GenerationEngine."""

from __future__ import annotations

import json
import os
import time
from typing import Dict, List, Optional

import torch

from ..data.tokenizer import ConversationTokenizer
from ..models.transformer import DeepSeekTransformer
from .engine import GenerationConfig, GenerationEngine, SAMPLING_MODES
from .loader import (find_latest_checkpoint, infer_config_from_state_dict,
                     load_checkpoint_smart)


class ChatInterface:
    def __init__(self, checkpoint: Optional[str] = None,
                 model: Optional[DeepSeekTransformer] = None,
                 tokenizer: Optional[ConversationTokenizer] = None,
                 device: Optional[str] = None,
                 history_window: int = 8,
                 system_prompt: str = "You are a helpful assistant.",
                 quantize: Optional[str] = None):
        self.device = torch.device(device) if device else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        if model is None:
            path = checkpoint or find_latest_checkpoint()
            if path is None:
                raise FileNotFoundError(
                    "no checkpoint found; pass checkpoint= or model=")
            payload = load_checkpoint_smart(path)
            sd = payload["model_state_dict"]
            cfg = infer_config_from_state_dict(sd)
            model = DeepSeekTransformer(cfg)
            model.load_state_dict(sd, strict=False)
            print(f"loaded {path}: {cfg.num_layers}L/{cfg.hidden_size}h, "
                  f"moe={cfg.use_moe}")
        self.model = model.to(self.device).eval()
        if quantize:
            from ..ops.quant import quantize_model, quantized_model_bytes
            n = quantize_model(self.model, mode=quantize)
            print(f"quantized {n} Linear layers to {quantize} "
                  f"({quantized_model_bytes(self.model) / 1e6:.1f} MB)")

        self.tokenizer = tokenizer or ConversationTokenizer()
        self.engine = GenerationEngine(self.model, self.tokenizer, self.device)
        self.gen_config = GenerationConfig.from_mode("standard")
        self.draft_model = None       # set_draft_model() enables speculative
        self.mode = "standard"
        self.system_prompt = system_prompt
        self.history: List[Dict] = []
        self.history_window = history_window
        self.session_stats = {"turns": 0, "start": time.time()}

    def set_draft_model(self, draft) -> None:
        """Enable speculative decoding: a small draft proposes, the main
        model verifies (exact greedy / distribution-preserving sampling —
        inference/engine.generate_speculative)."""
        self.draft_model = draft.to(self.device).eval()

    # ------------------------------------------------------------------
    def _prompt_ids(self, user_text: str) -> List[int]:
        msgs = [{"role": "system", "content": self.system_prompt}]
        msgs += self.history[-2 * self.history_window:]
        msgs.append({"role": "user", "content": user_text})
        ids: List[int] = []
        for m in msgs:
            ids.extend(self.tokenizer.encode_message(m["role"], m["content"]))
        # open an assistant turn for the model to complete
        ids += [self.tokenizer.special_tokens["<|im_start|>"],
                self.tokenizer.special_tokens["<|assistant|>"]]
        return ids

    def respond(self, user_text: str, stream: bool = False) -> str:
        ids = self._prompt_ids(user_text)
        pieces: List[str] = []

        def cb(tok):
            piece = self.tokenizer.decode([tok])
            pieces.append(piece)
            if stream:
                print(piece, end="", flush=True)

        if self.draft_model is not None:
            toks = self.engine.generate_speculative(ids, self.draft_model,
                                                    self.gen_config)
            for t in toks:     # speculative emits in verified blocks
                cb(t)
        else:
            self.engine.generate(ids, self.gen_config, stream_callback=cb)
        reply = "".join(pieces).strip()
        self.history.append({"role": "user", "content": user_text})
        self.history.append({"role": "assistant", "content": reply})
        self.session_stats["turns"] += 1
        return reply

    # ------------------------------------------------------------------
    def handle_command(self, line: str) -> Optional[str]:
        """Returns output text, or None if `line` is not a command."""
        if not line.startswith("/"):
            return None
        cmd, _, arg = line.partition(" ")
        if cmd == "/help":
            return ("commands: /help /stats /mode <name> /system <prompt> "
                    "/save <path> /config /clear /quit\n"
                    f"modes: {', '.join(SAMPLING_MODES)}")
        if cmd == "/stats":
            s = dict(self.engine.get_stats(), **self.session_stats)
            return json.dumps(s, indent=2, default=str)
        if cmd == "/mode":
            if arg not in SAMPLING_MODES:
                return f"unknown mode {arg!r}; choose from {list(SAMPLING_MODES)}"
            self.mode = arg
            keep = self.gen_config.max_new_tokens
            self.gen_config = GenerationConfig.from_mode(
                arg, max_new_tokens=keep)
            return f"sampling mode -> {arg}"
        if cmd == "/system":
            self.system_prompt = arg or self.system_prompt
            return "system prompt updated"
        if cmd == "/save":
            path = arg or f"chat_session_{int(time.time())}.json"
            with open(path, "w") as f:
                json.dump({"system": self.system_prompt,
                           "history": self.history}, f, indent=2)
            return f"saved to {path}"
        if cmd == "/config":
            return json.dumps(vars(self.gen_config), indent=2, default=str)
        if cmd == "/clear":
            self.history.clear()
            return "history cleared"
        if cmd in ("/quit", "/exit"):
            return "__QUIT__"
        return f"unknown command {cmd}; try /help"

    def run(self):
        print("LuminaAI-AMD chat — /help for commands, /quit to exit")
        while True:
            try:
                line = input("you> ").strip()
            except (EOFError, KeyboardInterrupt):
                break
            if not line:
                continue
            out = self.handle_command(line)
            if out == "__QUIT__":
                break
            if out is not None:
                print(out)
                continue
            print("ai> ", end="", flush=True)
            self.respond(line, stream=True)
            print()


def main():
    import argparse
    ap = argparse.ArgumentParser(description="LuminaAI-AMD chat REPL")
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--device", default=None)
    ap.add_argument("--quantize", default=None, choices=["int8", "int4", "fp8"],
                    help="weight-only quantization for inference")
    ap.add_argument("--draft-checkpoint", default=None,
                    help="small draft model checkpoint for speculative "
                         "decoding")
    args = ap.parse_args()
    chat = ChatInterface(checkpoint=args.checkpoint, device=args.device,
                         quantize=args.quantize)
    if args.draft_checkpoint:
        from ..models import DeepSeekTransformer
        from .loader import infer_config_from_state_dict, load_checkpoint_smart
        payload = load_checkpoint_smart(args.draft_checkpoint)
        dsd = payload["model_state_dict"]
        draft = DeepSeekTransformer(infer_config_from_state_dict(dsd))
        draft.load_state_dict(dsd, strict=False)
        chat.set_draft_model(draft)
        print(f"speculative decoding on (draft: {args.draft_checkpoint})")
    chat.run()


if __name__ == "__main__":
    main()
