/* Emit struct offsets of the LOCAL perl build as JSON.
 *
 * The agent's Perl unwinder (parca_agent_amd/interp/perl.py) walks the
 * interpreter's context stack in TARGET process memory; the offsets it
 * needs depend on the perl build configuration (ithreads, multiplicity),
 * so they are extracted here by compiling against the installed CORE
 * headers — the ground truth for this exact build — instead of shipping
 * per-version tables (reference analog: the fork's perl unwinder offset
 * tables). build_native.py compiles and runs this, storing the JSON next
 * to the package keyed by the perl binary's FileID.
 */
#include <stdio.h>
#include <stddef.h>

#include "EXTERN.h"
#include "perl.h"

int main(void) {
  printf("{\n");
  /* PerlInterpreter (struct interpreter) fields: with MULTIPLICITY the
   * PL_* vars are Ixxx members. */
  printf("  \"interp_curcop\": %zu,\n",
         offsetof(struct interpreter, Icurcop));
  printf("  \"interp_curstackinfo\": %zu,\n",
         offsetof(struct interpreter, Icurstackinfo));

  /* PERL_SI stack-info chain. */
  printf("  \"si_cxstack\": %zu,\n", offsetof(PERL_SI, si_cxstack));
  printf("  \"si_cxix\": %zu,\n", offsetof(PERL_SI, si_cxix));
  printf("  \"si_prev\": %zu,\n", offsetof(PERL_SI, si_prev));
  printf("  \"si_type\": %zu,\n", offsetof(PERL_SI, si_type));

  /* PERL_CONTEXT: type discriminator, caller COP, sub CV. */
  printf("  \"cx_size\": %zu,\n", sizeof(PERL_CONTEXT));
  printf("  \"cx_type\": %zu,\n",
         offsetof(PERL_CONTEXT, cx_u.cx_blk.blku_type));
  printf("  \"cx_oldcop\": %zu,\n",
         offsetof(PERL_CONTEXT, cx_u.cx_blk.blku_oldcop));
  printf("  \"cx_sub_cv\": %zu,\n",
         offsetof(PERL_CONTEXT, cx_u.cx_blk.blk_u.blku_sub.cv));
  printf("  \"cxtypemask\": %d,\n", CXTYPEMASK);
  printf("  \"cxt_sub\": %d,\n", CXt_SUB);
  printf("  \"cxt_format\": %d,\n", CXt_FORMAT);
  printf("  \"cxt_eval\": %d,\n", CXt_EVAL);

  /* COP source location. cop_file is char* under ithreads, GV* without;
   * the flag tells the reader which. */
  printf("  \"cop_line\": %zu,\n", offsetof(COP, cop_line));
  printf("  \"cop_file\": %zu,\n", offsetof(COP, cop_file));
#ifdef USE_ITHREADS
  printf("  \"cop_file_is_char\": 1,\n");
#else
  printf("  \"cop_file_is_char\": 0,\n");
#endif

  /* SV head + CV body: name via CvNAMED hek or CvGV gv. */
  printf("  \"sv_any\": %zu,\n", offsetof(SV, sv_any));
  printf("  \"sv_flags\": %zu,\n", offsetof(SV, sv_flags));
  printf("  \"xpvcv_gv_u\": %zu,\n", offsetof(XPVCV, xcv_gv_u));
  printf("  \"xpvcv_flags\": %zu,\n", offsetof(XPVCV, xcv_flags));
  printf("  \"cvf_named\": %d,\n", CVf_NAMED);

  /* GV name: ((XPVGV*)SvANY(gv))->xiv_u.xivu_namehek */
  printf("  \"xpvgv_namehek\": %zu,\n", offsetof(XPVGV, xiv_u));

  /* HEK layout. */
  printf("  \"hek_len\": %zu,\n", offsetof(HEK, hek_len));
  printf("  \"hek_key\": %zu\n", offsetof(HEK, hek_key));
  printf("}\n");
  return 0;
}
