from kfac_pytorch_amd.utils.harness import (LabelSmoothLoss, Metric,
                                            MultiEpochsDataLoader,
                                            PhaseTimers, create_lr_schedule,
                                            generate_pseudo_labels,
                                            load_checkpoint,
                                            polynomial_decay_lr,
                                            save_checkpoint)

__all__ = ["Metric", "LabelSmoothLoss", "PhaseTimers", "create_lr_schedule",
           "polynomial_decay_lr", "save_checkpoint", "load_checkpoint",
           "MultiEpochsDataLoader", "generate_pseudo_labels"]
