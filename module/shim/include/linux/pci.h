/* shim forward */
#include "rocnr_shim_all.h"
