/* filled in milestone 3 (BLS oracle) */
