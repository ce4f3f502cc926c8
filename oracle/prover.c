/* prover.c — placeholder for the CPU-oracle restatement of
 * halo2_proofs::plonk::create_proof (filled in as the round proceeds;
 * see DESIGN.md for the staged plan). ORACLE TEST INFRASTRUCTURE. */
