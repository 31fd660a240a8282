"""Model/run configuration (reference main.py:93-115 ``Option``).

The reference snapshots CLI args + vocab sizes into an Option object that is
the only thing the model layer sees; optuna mutates it in place.  Kept here
as a plain class with the same field names, plus MI355X-specific fields
(precision, dp rank/world) that have no reference counterpart.
"""

from __future__ import annotations


class Option:
    def __init__(
        self,
        terminal_count: int,
        path_count: int,
        label_count: int,
        max_path_length: int = 200,
        terminal_embed_size: int = 100,
        path_embed_size: int = 100,
        encode_size: int = 300,
        dropout_prob: float = 0.25,
        batch_size: int = 32,
        eval_method: str = "subtoken",
        angular_margin_loss: bool = False,
        angular_margin: float = 0.5,
        inverse_temp: float = 30.0,
        device=None,
        precision: str = "fp32",
    ) -> None:
        self.max_path_length = max_path_length
        self.terminal_count = terminal_count
        self.path_count = path_count
        self.label_count = label_count
        self.terminal_embed_size = terminal_embed_size
        self.path_embed_size = path_embed_size
        self.encode_size = encode_size
        self.dropout_prob = dropout_prob
        self.batch_size = batch_size
        self.eval_method = eval_method
        self.angular_margin_loss = angular_margin_loss
        self.angular_margin = angular_margin
        self.inverse_temp = inverse_temp
        self.device = device
        self.precision = precision

    @classmethod
    def from_args(cls, args, reader, device) -> "Option":
        return cls(
            terminal_count=len(reader.terminal_vocab),
            path_count=len(reader.path_vocab),
            label_count=len(reader.label_vocab),
            max_path_length=args.max_path_length,
            terminal_embed_size=args.terminal_embed_size,
            path_embed_size=args.path_embed_size,
            encode_size=args.encode_size,
            dropout_prob=args.dropout_prob,
            batch_size=args.batch_size,
            eval_method=args.eval_method,
            angular_margin_loss=args.angular_margin_loss,
            angular_margin=args.angular_margin,
            inverse_temp=args.inverse_temp,
            device=device,
            precision=getattr(args, "precision", "fp32"),
        )
