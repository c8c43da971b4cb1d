from greptimedb_amd.pipeline.engine import (Pipeline, PipelineError,
                                            PipelineStore)

__all__ = ["Pipeline", "PipelineError", "PipelineStore"]
