from tepdist_amd.data.synthetic import (SyntheticImages, SyntheticTokens,
                                        device_prefetcher)  # noqa: F401
