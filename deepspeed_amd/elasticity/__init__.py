from .elasticity import (ElasticityError, compute_elastic_config,
                         get_valid_gpus)

__all__ = ["compute_elastic_config", "get_valid_gpus", "ElasticityError"]
