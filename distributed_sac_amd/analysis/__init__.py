from .plot_utils import (attention_map, cal_z_context,  # noqa: F401
                         plot_attention_map, plot_cosine_similarity_map,
                         z_context_cosine_similarity)
