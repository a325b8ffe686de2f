from automodel_amd.eval.tool_calling import (  # noqa: F401
    METRIC_KEYS,
    ToolCall,
    ToolCallEvaluator,
    parse_tool_calls,
    score_tool_calls,
)
from automodel_amd.eval.perplexity import evaluate_nll  # noqa: F401
