from kfac_pytorch_amd.utils.harness import (LabelSmoothLoss, Metric,
                                            PhaseTimers, create_lr_schedule,
                                            load_checkpoint,
                                            polynomial_decay_lr,
                                            save_checkpoint)

__all__ = ["Metric", "LabelSmoothLoss", "PhaseTimers", "create_lr_schedule",
           "polynomial_decay_lr", "save_checkpoint", "load_checkpoint"]
