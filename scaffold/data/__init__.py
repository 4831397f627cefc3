from .synthetic import (SyntheticImageData, SyntheticTokenData,
                        SyntheticVectorData, build_data)

__all__ = ["SyntheticImageData", "SyntheticTokenData", "SyntheticVectorData", "build_data"]
