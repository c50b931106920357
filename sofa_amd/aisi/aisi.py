"""placeholder — implemented later this round"""
def sofa_aisi(logdir, cfg, df_cpu, df_gpu, df_rccl, features):
    raise NotImplementedError
