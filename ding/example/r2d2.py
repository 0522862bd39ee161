"""Memory-length R2D2 via the serial pipeline (recurrent unrolls; on MI355X
the LN-LSTM steps run through the fused HIP cell)."""
from ding.entry import serial_pipeline


def main(max_train_iter: int = 1000):
    from dizoo.memory.config.memory_len_r2d2_config import create_config, main_config
    cfg = (main_config, create_config)
    return serial_pipeline(cfg, seed=0, max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
