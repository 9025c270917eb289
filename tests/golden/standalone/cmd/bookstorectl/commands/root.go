/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package commands

import (
	"github.com/spf13/cobra"

	// common imports for subcommands
	cmdgenerate "github.com/acme/bookstore/cmd/bookstorectl/commands/generate"
	cmdinit "github.com/acme/bookstore/cmd/bookstorectl/commands/init"
	cmdversion "github.com/acme/bookstore/cmd/bookstorectl/commands/version"

	// specific imports for workloads
	generateapps "github.com/acme/bookstore/cmd/bookstorectl/commands/generate/apps"
	initapps "github.com/acme/bookstore/cmd/bookstorectl/commands/init/apps"
	versionapps "github.com/acme/bookstore/cmd/bookstorectl/commands/version/apps"
	//+operator-builder:subcommands:imports
)

// BookstorectlCommand represents the base command when called without any subcommands.
type BookstorectlCommand struct {
	*cobra.Command
}

// NewBookstorectlCommand returns an instance of the BookstorectlCommand.
func NewBookstorectlCommand() *BookstorectlCommand {
	c := &BookstorectlCommand{
		Command: &cobra.Command{
			Use:   "bookstorectl",
			Short: "Manage the bookstore application",
			Long:  "Manage the bookstore application",
		},
	}

	c.addSubCommands()

	return c
}

// Run represents the main entry point into the command
// This is called by main.main() to execute the root command.
func (c *BookstorectlCommand) Run() {
	cobra.CheckErr(c.Execute())
}

func (c *BookstorectlCommand) newInitSubCommand() {
	parentCommand := cmdinit.GetParent(c.Command)
	_ = parentCommand

	// add the init subcommands
	initapps.NewBookStoreSubCommand(parentCommand)
	//+operator-builder:subcommands:init
}

func (c *BookstorectlCommand) newGenerateSubCommand() {
	parentCommand := cmdgenerate.GetParent(c.Command)
	_ = parentCommand

	// add the generate subcommands
	generateapps.NewBookStoreSubCommand(parentCommand)
	//+operator-builder:subcommands:generate
}

func (c *BookstorectlCommand) newVersionSubCommand() {
	parentCommand := cmdversion.GetParent(c.Command)
	_ = parentCommand

	// add the version subcommands
	versionapps.NewBookStoreSubCommand(parentCommand)
	//+operator-builder:subcommands:version
}

// addSubCommands adds any additional subCommands to the root command.
func (c *BookstorectlCommand) addSubCommands() {
	c.newInitSubCommand()
	c.newGenerateSubCommand()
	c.newVersionSubCommand()
}
