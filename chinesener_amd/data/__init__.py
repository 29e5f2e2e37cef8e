"""L0/L1: corpora, tokenizers, word-enhance features, feature cache."""
