from .predict_utils import (decode_prediction, extract_entity, fix_tokens,
                            process_prediction)
from .entity_eval import (entity_report, extract_spans, tag_report,
                          report_to_text)

__all__ = ["decode_prediction", "extract_entity", "fix_tokens",
           "process_prediction", "entity_report", "extract_spans",
           "tag_report", "report_to_text"]
