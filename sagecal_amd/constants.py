"""Physical and tuning constants.

Mirrors the roles of /root/reference/src/lib/Dirac/Dirac_GPUtune.h and
scattered constants in Dirac_common.h, re-chosen for MI355X (gfx950).
"""

# speed of light [m/s] (reference: CONST_C in Dirac_common.h)
C_LIGHT = 299792458.0

# Earth angular velocity [rad/s] (reference: predict.c time_smear)
OMEGA_EARTH = 7.2921150e-5

# small epsilon used by smearing guards (reference: CLM_EPSILON)
CLM_EPSILON = 1e-12

# Student's-t defaults (reference: src/MS/data.cpp nulow/nuhigh defaults)
NU_LOW = 2.0
NU_HIGH = 30.0
NU_GRID = 30  # grid points for the AECM nu search (reference: Nd in update_nu callers)

# station-count threshold below which RTR solver modes fall back to LM and
# EM iterations are multiplied (reference: src/MS/sagecalmain.h LMCUT=40)
LMCUT = 40

# source types (reference: Dirac_common.h STYPE_*)
STYPE_POINT = 0
STYPE_GAUSSIAN = 1
STYPE_DISK = 2
STYPE_RING = 3
STYPE_SHAPELET = 4

# solver modes (reference: Dirac.h SM_* 1606-1613)
SM_LM_LBFGS = 0       # LM, no OS
SM_OSLM_LBFGS = 1     # OS accelerated LM
SM_OSLM_OSRLM_RLBFGS = 2  # OS robust LM
SM_RLM_RLBFGS = 3     # robust LM
SM_RTR_OSLM_LBFGS = 4  # RTR
SM_RTR_OSRLM_RLBFGS = 5  # robust RTR (reference default -j 5)
SM_NSD_RLBFGS = 6     # Nesterov SD

ROBUST_MODES = {SM_OSLM_OSRLM_RLBFGS, SM_RLM_RLBFGS, SM_RTR_OSRLM_RLBFGS, SM_NSD_RLBFGS}
