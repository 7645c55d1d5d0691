from typing import Any, Union


class Preprocess(object):
    """Tokenize text for the BERT endpoint (equivalent of the reference's
    huggingface example preprocess)."""

    def __init__(self):
        self.model_endpoint = None
        self._tok = None

    def load(self, local_file_name: str) -> Any:
        return None  # model built by the engine from the model card

    def _tokenizer(self):
        if self._tok is None:
            from tokenizers import BertWordPieceTokenizer

            self._tok = BertWordPieceTokenizer()
        return self._tok

    def preprocess(self, body: Union[bytes, dict], state: dict,
                   collect_custom_statistics_fn=None) -> Any:
        # accept pre-tokenized ids, or raw text when a vocab is available
        if isinstance(body, dict) and "input_ids" in body:
            ids = body["input_ids"]
            mask = body.get("attention_mask", [1] * len(ids))
            return {"input_ids": ids, "attention_mask": mask}
        text = body.get("text", "") if isinstance(body, dict) else str(body)
        enc = self._tokenizer().encode(text)
        return {"input_ids": enc.ids, "attention_mask": enc.attention_mask}

    def postprocess(self, data: Any, state: dict,
                    collect_custom_statistics_fn=None) -> dict:
        import numpy as np

        logits = np.asarray(data)
        return {"label": int(logits.argmax()), "logits": logits.tolist()}
